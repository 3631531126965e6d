/* test_config_seqlock.c — cross-process seqlock consistency.
 *
 * A writer process continuously mutates a device_t pair of fields that
 * must be observed together (total_memory and core_limit written as a
 * matched pair); reader processes snapshot via the seqlock protocol and
 * assert they never observe a torn pair.  Mirrors the Go-writer /
 * C-reader runtime-mutation path (reference resource_data seqlock
 * design doc).
 */
#define _GNU_SOURCE
#include "../../include/hook.h"
#include "../../include/shm.h"

#include <assert.h>
#include <sched.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/wait.h>
#include <unistd.h>

#define NREADERS 6
#define WRITES 50000
#define READS 200000

/* the library's snapshot read (hip_hook.c get_device_snapshot)        */
static void snapshot(const device_t *dev, device_t *out) {
    for (;;) {
        uint32_t s0 = seq_load(&dev->seq);
        if (s0 & 1u) {
            sched_yield();
            continue;
        }
        memcpy(out, dev, sizeof(*out));
        if (seq_read_valid(&dev->seq, s0)) return;
    }
}

int main(void) {
    char path[256];
    snprintf(path, sizeof(path), "/tmp/vgpu_test_seqlock_%d.bin",
             (int)getpid());
    bool created;
    resource_data_t *r = vgpu_region_attach(path, sizeof(resource_data_t),
                                            VGPU_CFG_MAGIC, true, &created);
    assert(r && created);
    device_t *dev = &r->devices[0];
    dev->total_memory = 1000;
    dev->core_limit = 1;

    pid_t kids[NREADERS];
    for (int i = 0; i < NREADERS; i++) {
        pid_t pid = fork();
        assert(pid >= 0);
        if (pid == 0) {
            resource_data_t *cr = vgpu_region_attach(
                path, sizeof(resource_data_t), VGPU_CFG_MAGIC, false, NULL);
            assert(cr);
            device_t snap;
            for (int k = 0; k < READS; k++) {
                snapshot(&cr->devices[0], &snap);
                /* invariant: total_memory == core_limit * 1000        */
                if (snap.total_memory != (uint64_t)snap.core_limit * 1000) {
                    fprintf(stderr, "FAIL: torn read tm=%llu cl=%u\n",
                            (unsigned long long)snap.total_memory,
                            snap.core_limit);
                    _exit(1);
                }
            }
            _exit(0);
        }
        kids[i] = pid;
    }

    /* writer: bump the pair under the seqlock                          */
    for (uint32_t w = 2; w < WRITES; w++) {
        seq_write_begin(&dev->seq);
        dev->core_limit = w;
        dev->total_memory = (uint64_t)w * 1000;
        seq_write_end(&dev->seq);
    }

    int fail = 0;
    for (int i = 0; i < NREADERS; i++) {
        int st;
        waitpid(kids[i], &st, 0);
        if (!WIFEXITED(st) || WEXITSTATUS(st) != 0) fail = 1;
    }
    unlink(path);
    if (!fail)
        printf("PASS seqlock: %d readers x %d reads, %d writes\n", NREADERS,
               READS, WRITES);
    return fail;
}
