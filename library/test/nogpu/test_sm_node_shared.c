/* test_sm_node_shared.c — multi-process token-bucket conservation.
 *
 * Forks N worker processes onto one MAP_SHARED sm_node region.  Each
 * worker performs K CAS token decrements (the launch-path operation)
 * while the parent concurrently refills (the watcher operation).  At
 * the end: initial + total_refilled - total_consumed == final tokens,
 * i.e. no token is lost or double-counted under contention.
 *
 * Parity: reference library/test/nogpu/test_sm_node_shared.c concept —
 * this is how cross-process correctness is proven without a GPU.
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
#include <time.h>
#include <unistd.h>

#define NPROC 8
#define NITER 200000
#define GRIDS 7 /* tokens per launch */

static char g_path[256];

/* the launch-path decrement: identical logic to rate_limiter's CAS    */
static void consume(sm_node_dev_t *dev, int64_t want) {
    for (;;) {
        int64_t cur = __atomic_load_n(&dev->tokens, __ATOMIC_RELAXED);
        if (cur < 0) {
            /* throttled: in the library we sleep TIME_TICK; here yield */
            sched_yield();
            continue;
        }
        if (__atomic_compare_exchange_n(&dev->tokens, &cur, cur - want,
                                        true, __ATOMIC_ACQ_REL,
                                        __ATOMIC_RELAXED))
            return;
    }
}

int main(void) {
    snprintf(g_path, sizeof(g_path), "/tmp/vgpu_test_smnode_%d.bin",
             (int)getpid());
    bool created = false;
    sm_node_region_t *r = vgpu_region_attach(
        g_path, sizeof(sm_node_region_t), VGPU_SMND_MAGIC, true, &created);
    assert(r && created);
    sm_node_dev_t *dev = &r->devices[0];
    const int64_t pool = (int64_t)NPROC * NITER * GRIDS * 2;
    __atomic_store_n(&dev->pool_size, pool, __ATOMIC_RELAXED);
    __atomic_store_n(&dev->tokens, pool, __ATOMIC_RELAXED);

    pid_t kids[NPROC];
    for (int i = 0; i < NPROC; i++) {
        pid_t pid = fork();
        assert(pid >= 0);
        if (pid == 0) {
            /* child: fresh attach (not inherited mapping) to exercise
             * the attach path cross-process */
            sm_node_region_t *cr = vgpu_region_attach(
                g_path, sizeof(sm_node_region_t), VGPU_SMND_MAGIC, false,
                NULL);
            assert(cr);
            for (int k = 0; k < NITER; k++) consume(&cr->devices[0], GRIDS);
            _exit(0);
        }
        kids[i] = pid;
    }

    /* parent: concurrent refills (watcher role), conserving total     */
    int64_t refilled = 0;
    for (int c = 0; c < 1000; c++) {
        int64_t add = 1000;
        __atomic_fetch_add(&dev->tokens, add, __ATOMIC_ACQ_REL);
        refilled += add;
        struct timespec ts = {0, 100000};
        nanosleep(&ts, NULL);
    }

    for (int i = 0; i < NPROC; i++) {
        int st;
        waitpid(kids[i], &st, 0);
        assert(WIFEXITED(st) && WEXITSTATUS(st) == 0);
    }

    int64_t final = __atomic_load_n(&dev->tokens, __ATOMIC_SEQ_CST);
    int64_t consumed = (int64_t)NPROC * NITER * GRIDS;
    int64_t expect = pool + refilled - consumed;
    if (final != expect) {
        fprintf(stderr, "FAIL: final=%lld expect=%lld\n", (long long)final,
                (long long)expect);
        return 1;
    }
    vgpu_region_detach(r, sizeof(sm_node_region_t));
    unlink(g_path);
    printf("PASS token conservation: %d procs x %d iters\n", NPROC, NITER);
    return 0;
}
