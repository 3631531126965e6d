/* sq_sampler — standalone host-side utilization watcher for gfx950.
 *
 * The high-fidelity source for the shared SM-utilization region
 * (sm_util.config): per-device GPU busy sampled from the hardware
 * counters (GRBM_GUI_ACTIVE/GRBM_COUNT via rocprofiler-sdk's device
 * counting service — what `rocprofv3 --pmc` reads) plus per-process
 * gfx engine time and VRAM from amd-smi.  Containers' shims mmap the
 * region read-only instead of each paying their own query cost
 * (reference: the SharedSMUtilizationWatcher architecture;
 * pkg/device/manager/watcher.go + hook.h:561-573 rationale).
 *
 * One instance per node (run by device-monitor or the device-plugin
 * pod).  Usage:
 *   sq_sampler [--out /etc/vgpu-manager/watcher/sm_util.config]
 *              [--interval-ms 80] [--once]
 *
 * rocprofiler usage follows the public AMD sample
 * (rocprofiler-sdk samples, device_counting_sync_client.cpp, MIT).
 */
#define __HIP_PLATFORM_AMD__ 1
#include <hip/hip_runtime_api.h>

#include <rocprofiler-sdk/registration.h>
#include <rocprofiler-sdk/rocprofiler.h>

#include <amd_smi/amdsmi.h>

extern "C" {
#include "../include/hook.h"
#include "../include/shm.h"
}

#include <atomic>
#include <chrono>
#include <cstring>
#include <dirent.h>
#include <functional>
#include <map>
#include <string>
#include <thread>
#include <vector>

#include <stdio.h>
#include <time.h>
#include <unistd.h>

static uint64_t mono_ns_now(void) {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return (uint64_t)ts.tv_sec * 1000000000ull + (uint64_t)ts.tv_nsec;
}

#define RP_CHECK(x)                                                     \
    do {                                                                \
        rocprofiler_status_t st_ = (x);                                 \
        if (st_ != ROCPROFILER_STATUS_SUCCESS) {                        \
            fprintf(stderr, "sq_sampler: %s failed: %s\n", #x,          \
                    rocprofiler_get_status_string(st_));                \
            return false;                                               \
        }                                                               \
    } while (0)

/* ---- one counter-sampling context per GPU agent ------------------- */
struct AgentSampler {
    rocprofiler_agent_id_t agent{};
    rocprofiler_context_id_t ctx{};
    rocprofiler_buffer_id_t buf{};
    rocprofiler_counter_config_id_t profile{};
    size_t record_cap = 0;
    /* counter-id -> name (GPU_UTIL / SQ_WAVES) */
    std::map<uint64_t, std::string> names;

    bool init(rocprofiler_agent_id_t a);
    bool sample(uint32_t window_ms, double *gpu_util_pct,
                double *sq_waves,
                const std::function<void()> &tick = nullptr);
};

static std::vector<rocprofiler_agent_v0_t> g_agents;
static std::vector<AgentSampler> g_samplers;
static std::atomic<int> g_tool_ready{0};

static bool make_profile(AgentSampler *s) {
    struct Ctx {
        std::vector<rocprofiler_counter_id_t> want;
        std::map<uint64_t, std::string> *names;
    } c{{}, &s->names};
    auto cb = [](rocprofiler_agent_id_t, rocprofiler_counter_id_t *ids,
                 size_t n, void *ud) {
        auto *cc = static_cast<Ctx *>(ud);
        for (size_t i = 0; i < n; i++) {
            rocprofiler_counter_info_v0_t info;
            if (rocprofiler_query_counter_info(
                    ids[i], ROCPROFILER_COUNTER_INFO_VERSION_0,
                    &info) != ROCPROFILER_STATUS_SUCCESS)
                continue;
            if (strcmp(info.name, "GPU_UTIL") == 0 ||
                strcmp(info.name, "SQ_WAVES") == 0) {
                cc->want.push_back(ids[i]);
                (*cc->names)[ids[i].handle] = info.name;
            }
        }
        return ROCPROFILER_STATUS_SUCCESS;
    };
    RP_CHECK(rocprofiler_iterate_agent_supported_counters(s->agent, cb,
                                                          &c));
    if (c.want.empty()) {
        fprintf(stderr, "sq_sampler: no GPU_UTIL/SQ_WAVES on agent\n");
        return false;
    }
    RP_CHECK(rocprofiler_create_counter_config(
        s->agent, c.want.data(), c.want.size(), &s->profile));
    s->record_cap = 0;
    for (auto id : c.want) {
        rocprofiler_counter_info_v1_t info;
        if (rocprofiler_query_counter_info(
                id, ROCPROFILER_COUNTER_INFO_VERSION_1, &info) ==
            ROCPROFILER_STATUS_SUCCESS)
            s->record_cap += info.dimensions_instances_count;
        else
            s->record_cap += 1024;
    }
    return true;
}

bool AgentSampler::init(rocprofiler_agent_id_t a) {
    agent = a;
    RP_CHECK(rocprofiler_create_context(&ctx));
    RP_CHECK(rocprofiler_create_buffer(
        ctx, 4096, 2048, ROCPROFILER_BUFFER_POLICY_LOSSLESS,
        [](rocprofiler_context_id_t, rocprofiler_buffer_id_t,
           rocprofiler_record_header_t **, size_t, void *, uint64_t) {},
        nullptr, &buf));
    rocprofiler_callback_thread_t thr{};
    RP_CHECK(rocprofiler_create_callback_thread(&thr));
    RP_CHECK(rocprofiler_assign_callback_thread(buf, thr));
    RP_CHECK(rocprofiler_configure_device_counting_service(
        ctx, buf, agent,
        [](rocprofiler_context_id_t c, rocprofiler_agent_id_t,
           rocprofiler_device_counting_agent_cb_t set_config,
           void *ud) {
            auto *self = static_cast<AgentSampler *>(ud);
            if (self->profile.handle != 0) set_config(c, self->profile);
        },
        this));
    return make_profile(this);
}

bool AgentSampler::sample(uint32_t window_ms, double *gpu_util_pct,
                          double *sq_waves,
                          const std::function<void()> &tick) {
    std::vector<rocprofiler_counter_record_t> recs(record_cap);
    RP_CHECK(rocprofiler_start_context(ctx));
    /* ~10ms ticks during the counting window: callers sub-sample
     * per-process KFD occupancy for an unbiased duty mean            */
    for (uint32_t t = 0; t < window_ms; t += 10) {
        std::this_thread::sleep_for(std::chrono::milliseconds(
            window_ms - t > 10 ? 10 : window_ms - t));
        if (tick) tick();
    }
    size_t n = recs.size();
    rocprofiler_status_t st = rocprofiler_sample_device_counting_service(
        ctx, {}, ROCPROFILER_COUNTER_FLAG_NONE, recs.data(), &n);
    rocprofiler_stop_context(ctx);
    if (st != ROCPROFILER_STATUS_SUCCESS) return false;
    double util = 0, waves = 0;
    for (size_t i = 0; i < n; i++) {
        rocprofiler_counter_id_t cid{};
        rocprofiler_query_record_counter_id(recs[i].id, &cid);
        auto it = names.find(cid.handle);
        if (it == names.end()) continue;
        if (it->second == "GPU_UTIL")
            util = util > recs[i].counter_value ? util
                                                : recs[i].counter_value;
        else
            waves += recs[i].counter_value;
    }
    *gpu_util_pct = util;
    *sq_waves = waves;
    return true;
}

/* ---- rocprofiler tool registration -------------------------------- */
static int tool_init(rocprofiler_client_finalize_t, void *) {
    auto cb = [](rocprofiler_agent_version_t ver, const void **arr,
                 size_t n, void *) {
        if (ver != ROCPROFILER_AGENT_INFO_VERSION_0)
            return ROCPROFILER_STATUS_ERROR;
        for (size_t i = 0; i < n; i++) {
            const auto *ag =
                static_cast<const rocprofiler_agent_v0_t *>(arr[i]);
            if (ag->type == ROCPROFILER_AGENT_TYPE_GPU)
                g_agents.push_back(*ag);
        }
        return ROCPROFILER_STATUS_SUCCESS;
    };
    if (rocprofiler_query_available_agents(
            ROCPROFILER_AGENT_INFO_VERSION_0, cb,
            sizeof(rocprofiler_agent_t),
            nullptr) != ROCPROFILER_STATUS_SUCCESS)
        return -1;
    g_samplers.resize(g_agents.size());
    for (size_t i = 0; i < g_agents.size(); i++)
        if (!g_samplers[i].init(g_agents[i].id)) return -1;
    g_tool_ready.store(1);
    return 0;
}

extern "C" rocprofiler_tool_configure_result_t *rocprofiler_configure(
    uint32_t version, const char *, uint32_t, rocprofiler_client_id_t *id) {
    (void)version;
    id->name = "vgpu-sq-sampler";
    static rocprofiler_tool_configure_result_t result{
        sizeof(rocprofiler_tool_configure_result_t), &tool_init,
        nullptr, nullptr};
    return &result;
}

/* ---- per-pid KFD occupancy sub-sampling --------------------------- */
/* kfd proc stats cu_occupancy files: CUs the
 * process occupies RIGHT NOW.  Point samples -> accumulate over the
 * window, publish the MEAN (same estimator the in-container shim
 * uses; full-or-zero point samples whipsaw controllers).            */
struct KfdOccWindow {
    std::map<int, std::pair<uint64_t, uint32_t>> acc; /* pid -> sum,n */

    void tick() {
        DIR *d = opendir("/sys/class/kfd/kfd/proc");
        if (!d) return;
        struct dirent *e;
        while ((e = readdir(d)) != nullptr) {
            if (e->d_name[0] < '0' || e->d_name[0] > '9') continue;
            int pid = atoi(e->d_name);
            char pdir[320];
            snprintf(pdir, sizeof(pdir),
                     "/sys/class/kfd/kfd/proc/%s", e->d_name);
            DIR *pd = opendir(pdir);
            if (!pd) continue;
            uint64_t occ = 0;
            struct dirent *se;
            while ((se = readdir(pd)) != nullptr) {
                if (strncmp(se->d_name, "stats_", 6) != 0) continue;
                char fp[640];
                snprintf(fp, sizeof(fp), "%s/%s/cu_occupancy", pdir,
                         se->d_name);
                FILE *f = fopen(fp, "re");
                if (!f) continue;
                unsigned v = 0;
                if (fscanf(f, "%u", &v) == 1) occ += v;
                fclose(f);
            }
            closedir(pd);
            auto &a = acc[pid];
            a.first += occ;
            a.second++;
        }
        closedir(d);
    }

    uint32_t mean(int pid) const {
        auto it = acc.find(pid);
        if (it == acc.end() || it->second.second == 0) return 0;
        return (uint32_t)(it->second.first / it->second.second);
    }
};

/* ---- amd-smi per-process view ------------------------------------- */
struct SmiState {
    bool ok = false;
    std::vector<amdsmi_processor_handle> handles;
    /* pid -> previous gfx engine ns (per device) */
    std::vector<std::map<int, uint64_t>> prev_gfx;
    uint64_t prev_ns = 0;

    bool init();
    void fill(int dev, device_util_t *out, uint64_t now,
              const KfdOccWindow *occ = nullptr);
};

bool SmiState::init() {
    if (amdsmi_init(AMDSMI_INIT_AMD_GPUS) != AMDSMI_STATUS_SUCCESS)
        return false;
    uint32_t nsock = 0;
    if (amdsmi_get_socket_handles(&nsock, nullptr) !=
        AMDSMI_STATUS_SUCCESS)
        return false;
    std::vector<amdsmi_socket_handle> socks(nsock);
    amdsmi_get_socket_handles(&nsock, socks.data());
    for (auto s : socks) {
        uint32_t np = 0;
        if (amdsmi_get_processor_handles(s, &np, nullptr) !=
            AMDSMI_STATUS_SUCCESS)
            continue;
        std::vector<amdsmi_processor_handle> ph(np);
        amdsmi_get_processor_handles(s, &np, ph.data());
        for (auto h : ph) handles.push_back(h);
    }
    prev_gfx.resize(handles.size());
    ok = !handles.empty();
    return ok;
}

void SmiState::fill(int dev, device_util_t *out, uint64_t now,
                    const KfdOccWindow *occ) {
    if (!ok || dev >= (int)handles.size()) return;
    amdsmi_processor_handle h = handles[dev];
    uint64_t vram = 0;
    if (amdsmi_get_gpu_memory_usage(h, AMDSMI_MEM_TYPE_VRAM, &vram) ==
        AMDSMI_STATUS_SUCCESS)
        out->vram_used_bytes = vram;
    uint32_t n = MAX_UTIL_PROCS;
    std::vector<amdsmi_proc_info_t> procs(MAX_UTIL_PROCS);
    if (amdsmi_get_gpu_process_list(h, &n, procs.data()) !=
        AMDSMI_STATUS_SUCCESS)
        return;
    if (n > MAX_UTIL_PROCS) n = MAX_UTIL_PROCS;
    double dt = prev_ns ? (double)(now - prev_ns) : 0;
    uint32_t count = 0;
    for (uint32_t i = 0; i < n && count < MAX_UTIL_PROCS; i++) {
        int pid = (int)procs[i].pid;
        uint64_t gfx = procs[i].engine_usage.gfx; /* ns cumulative */
        uint32_t permille = 0;
        auto it = prev_gfx[dev].find(pid);
        if (it != prev_gfx[dev].end() && dt > 0 && gfx >= it->second)
            permille =
                (uint32_t)((double)(gfx - it->second) * 1000.0 / dt);
        prev_gfx[dev][pid] = gfx;
        out->procs[count].pid = pid;
        out->procs[count].gfx_busy_permille =
            permille > 1000 ? 1000 : permille;
        out->procs[count].vram_bytes = procs[i].memory_usage.vram_mem;
        /* window-mean occupancy when available (unbiased duty),
         * else amd-smi's point sample                               */
        uint32_t m = occ ? occ->mean(pid) : 0;
        out->procs[count].cu_occupancy =
            m ? m : procs[i].cu_occupancy;
        count++;
    }
    out->proc_count = count;
}

/* ---- main ---------------------------------------------------------- */
int main(int argc, char **argv) {
    const char *out_path = VGPU_UTIL_PATH;
    uint32_t interval_ms = 80;
    bool once = false;
    for (int i = 1; i < argc; i++) {
        if (!strcmp(argv[i], "--out") && i + 1 < argc)
            out_path = argv[++i];
        else if (!strcmp(argv[i], "--interval-ms") && i + 1 < argc)
            interval_ms = (uint32_t)atoi(argv[++i]);
        else if (!strcmp(argv[i], "--once"))
            once = true;
        else {
            fprintf(stderr,
                    "usage: sq_sampler [--out PATH] [--interval-ms N] "
                    "[--once]\n");
            return 2;
        }
    }

    /* register BEFORE runtime init so the counting service attaches */
    rocprofiler_force_configure(&rocprofiler_configure);
    if (hipInit(0) != hipSuccess) {
        fprintf(stderr, "sq_sampler: hipInit failed (no GPU?)\n");
        return 1;
    }
    if (!g_tool_ready.load()) {
        fprintf(stderr, "sq_sampler: rocprofiler tool init failed\n");
        return 1;
    }

    util_region_t *region = (util_region_t *)vgpu_region_attach(
        out_path, sizeof(util_region_t), VGPU_UTIL_MAGIC,
        /*create=*/true, NULL);
    if (!region) {
        fprintf(stderr, "sq_sampler: cannot map %s\n", out_path);
        return 1;
    }

    SmiState smi;
    smi.init();

    int ndev = (int)g_samplers.size();
    if (ndev > MAX_DEVICE_COUNT) ndev = MAX_DEVICE_COUNT;
    region->device_count = (uint32_t)ndev;
    fprintf(stderr, "sq_sampler: %d GPUs -> %s every %ums (smi %s)\n",
            ndev, out_path, interval_ms, smi.ok ? "on" : "off");

    for (;;) {
        uint64_t now = mono_ns_now();
        for (int d = 0; d < ndev; d++) {
            double util = 0, waves = 0;
            KfdOccWindow occ;
            /* the counting window IS the cadence: GPU_UTIL =
             * GRBM_GUI_ACTIVE/GRBM_COUNT over the window; per-pid
             * occupancy is sub-sampled every ~10ms inside it */
            if (!g_samplers[d].sample(interval_ms, &util, &waves,
                                      [&occ] { occ.tick(); }))
                continue;
            device_util_t *u = &region->devices[d];
            __atomic_fetch_add(&u->seq, 1, __ATOMIC_ACQ_REL); /* odd */
            u->dev_busy_permille = (uint32_t)(util * 10.0);
            u->sample_ns = now;
            smi.fill(d, u, now, &occ);
            __atomic_fetch_add(&u->seq, 1, __ATOMIC_RELEASE); /* even */
        }
        smi.prev_ns = now;
        __atomic_store_n(&region->heartbeat_ns, now, __ATOMIC_RELEASE);
        if (once) break;
    }
    return 0;
}
