/* mem_view_tool — prints the container's vGPU quota view from the
 * mmap'd regions (debug aid; reference library/tools/mem_view_tool). */
#define _GNU_SOURCE
#include "../include/hook.h"
#include "../include/shm.h"

#include <stdio.h>
#include <stdlib.h>

int main(int argc, char **argv) {
    const char *cfg_path = argc > 1 ? argv[1] : VGPU_CONFIG_PATH;
    const char *vmem_path = argc > 2 ? argv[2] : VGPU_VMEM_PATH;
    resource_data_t *cfg = vgpu_region_attach(
        cfg_path, sizeof(resource_data_t), VGPU_CFG_MAGIC, false, NULL);
    if (!cfg) {
        fprintf(stderr, "no vgpu.config at %s\n", cfg_path);
        return 1;
    }
    printf("pod=%s/%s container=%s devices=%d policy=%u oversold=%u\n",
           cfg->pod_namespace, cfg->pod_name, cfg->container_name,
           cfg->device_count, cfg->compute_policy, cfg->oversold);
    vmem_region_t *vm = vgpu_region_attach(
        vmem_path, sizeof(vmem_region_t), VGPU_VMEM_MAGIC, false, NULL);
    for (int i = 0; i < cfg->device_count; i++) {
        device_t *d = &cfg->devices[i];
        unsigned long long hooked = 0, vmem = 0;
        if (vm) {
            hooked = (unsigned long long)__atomic_load_n(
                &vm->counters[i].dev_hooked_used, __ATOMIC_ACQUIRE);
            vmem = (unsigned long long)__atomic_load_n(
                &vm->counters[i].vmem_used, __ATOMIC_ACQUIRE);
        }
        printf("dev %d host=%d uuid=%.48s flags=%#x quota=%llu MiB "
               "cores=%u soft=%u hooked_used=%llu MiB vmem=%llu MiB\n",
               i, d->host_index, d->uuid, d->flags,
               (unsigned long long)(d->total_memory >> 20), d->core_limit,
               d->soft_core_limit, hooked >> 20, vmem >> 20);
    }
    return 0;
}
