/* test_vmem_region_concurrency.c — ledger consistency under fork storm.
 *
 * N processes concurrently claim ledger records (CAS FREE->BUSY->LIVE),
 * charge the per-device counters, then retire their own records.  At
 * the end every counter must be exactly zero and every record FREE —
 * proving no slot is double-claimed and no byte leaks cross-process.
 */
#define _GNU_SOURCE
#include "../../include/hook.h"
#include "../../include/shm.h"

#include <assert.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/wait.h>
#include <unistd.h>

#define NPROC 8
#define NITER 5000
#define LIVE_MAX 32 /* live records held per process at a time */

static int claim_record(vmem_region_t *r, uint64_t dptr, uint64_t size,
                        int dev) {
    for (uint32_t i = 0; i < MAX_VMEM_RECORDS; i++) {
        uint32_t st = VMEM_STATE_FREE;
        if (__atomic_compare_exchange_n(&r->records[i].state, &st,
                                        VMEM_STATE_BUSY, true,
                                        __ATOMIC_ACQ_REL, __ATOMIC_RELAXED)) {
            vmem_record_t *rec = &r->records[i];
            rec->kind = VMEM_KIND_SYNC;
            rec->dptr = dptr;
            rec->size = size;
            rec->pid = (int32_t)getpid();
            rec->device = dev;
            rec->created_ns = mono_ns();
            __atomic_fetch_add(&r->counters[dev].vmem_used, size,
                               __ATOMIC_ACQ_REL);
            __atomic_store_n(&rec->state, VMEM_STATE_LIVE, __ATOMIC_RELEASE);
            return (int)i;
        }
    }
    return -1;
}

static void retire_record(vmem_region_t *r, int idx) {
    vmem_record_t *rec = &r->records[idx];
    uint32_t st = VMEM_STATE_LIVE;
    int ok = __atomic_compare_exchange_n(&rec->state, &st, VMEM_STATE_BUSY,
                                         true, __ATOMIC_ACQ_REL,
                                         __ATOMIC_RELAXED);
    assert(ok);
    __atomic_fetch_sub(&r->counters[rec->device].vmem_used, rec->size,
                       __ATOMIC_ACQ_REL);
    __atomic_store_n(&rec->state, VMEM_STATE_FREE, __ATOMIC_RELEASE);
}

int main(void) {
    char path[256];
    snprintf(path, sizeof(path), "/tmp/vgpu_test_vmem_%d.bin", (int)getpid());
    bool created = false;
    vmem_region_t *r = vgpu_region_attach(path, sizeof(vmem_region_t),
                                          VGPU_VMEM_MAGIC, true, &created);
    assert(r && created);
    r->record_cap = MAX_VMEM_RECORDS;

    pid_t kids[NPROC];
    for (int i = 0; i < NPROC; i++) {
        pid_t pid = fork();
        assert(pid >= 0);
        if (pid == 0) {
            vmem_region_t *cr = vgpu_region_attach(
                path, sizeof(vmem_region_t), VGPU_VMEM_MAGIC, false, NULL);
            assert(cr);
            int live[LIVE_MAX];
            int nlive = 0;
            unsigned seed = (unsigned)getpid();
            for (int k = 0; k < NITER; k++) {
                if (nlive < LIVE_MAX && (rand_r(&seed) & 1)) {
                    int dev = rand_r(&seed) % 4;
                    int idx = claim_record(cr, 0x1000u * (unsigned)k,
                                           4096u * (unsigned)(dev + 1), dev);
                    assert(idx >= 0);
                    live[nlive++] = idx;
                } else if (nlive > 0) {
                    retire_record(cr, live[--nlive]);
                }
            }
            while (nlive > 0) retire_record(cr, live[--nlive]);
            _exit(0);
        }
        kids[i] = pid;
    }
    for (int i = 0; i < NPROC; i++) {
        int st;
        waitpid(kids[i], &st, 0);
        assert(WIFEXITED(st) && WEXITSTATUS(st) == 0);
    }

    int fail = 0;
    for (int d = 0; d < MAX_DEVICE_COUNT; d++) {
        uint64_t used = __atomic_load_n(&r->counters[d].vmem_used,
                                        __ATOMIC_SEQ_CST);
        if (used != 0) {
            fprintf(stderr, "FAIL: dev %d vmem_used=%llu\n", d,
                    (unsigned long long)used);
            fail = 1;
        }
    }
    for (uint32_t i = 0; i < MAX_VMEM_RECORDS; i++) {
        if (r->records[i].state != VMEM_STATE_FREE) {
            fprintf(stderr, "FAIL: record %u state=%u\n", i,
                    r->records[i].state);
            fail = 1;
        }
    }
    unlink(path);
    if (!fail) printf("PASS vmem ledger: %d procs x %d iters\n", NPROC, NITER);
    return fail;
}
