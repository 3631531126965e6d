/* mem_occupy_tool — allocates and holds device memory (workload
 * generator for quota tests; reference library/tools/mem_occupy_tool).
 * Usage: mem_occupy_tool <bytes> [seconds]                            */
#define _GNU_SOURCE
#include <dlfcn.h>
#include <stdio.h>
#include <stdlib.h>
#include <unistd.h>

typedef int (*malloc_fn)(void **, size_t);
typedef int (*free_fn)(void *);
typedef int (*info_fn)(size_t *, size_t *);

int main(int argc, char **argv) {
    if (argc < 2) {
        fprintf(stderr, "usage: %s <bytes> [hold_seconds]\n", argv[0]);
        return 2;
    }
    size_t bytes = (size_t)strtoull(argv[1], NULL, 0);
    int hold = argc > 2 ? atoi(argv[2]) : 3600;
    void *h = dlopen("libamdhip64.so.7", RTLD_LAZY);
    if (!h) h = dlopen("libamdhip64.so", RTLD_LAZY);
    if (!h) {
        fprintf(stderr, "cannot load libamdhip64\n");
        return 1;
    }
    malloc_fn hip_malloc = (malloc_fn)dlsym(h, "hipMalloc");
    info_fn hip_info = (info_fn)dlsym(h, "hipMemGetInfo");
    if (!hip_malloc) return 1;
    void *p = NULL;
    int rc = hip_malloc(&p, bytes);
    if (rc != 0) {
        fprintf(stderr, "hipMalloc(%zu) failed rc=%d\n", bytes, rc);
        return 1;
    }
    size_t freeb = 0, total = 0;
    if (hip_info) hip_info(&freeb, &total);
    printf("holding %zu bytes (view: free=%zu total=%zu) for %ds\n",
           bytes, freeb, total, hold);
    fflush(stdout);
    sleep((unsigned)hold);
    return 0;
}
