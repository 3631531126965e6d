/* virt_mem_tool — dumps the vmem ledger records (managed/host spill
 * past the HBM quota) from the shared vmem_node region (debug aid;
 * reference library/tools/virt_mem_tool). */
#define _GNU_SOURCE
#include "../include/hook.h"
#include "../include/shm.h"

#include <stdio.h>
#include <stdlib.h>

static const char *kind_name(uint32_t k) {
    switch (k) {
    case VMEM_KIND_SYNC:          return "UVA_SYNC";
    case VMEM_KIND_ASYNC:         return "UVA_ASYNC";
    case VMEM_KIND_CAPTURE:       return "CAPTURE";
    case VMEM_KIND_ASYNC_BRIDGE:  return "ASYNC_BRIDGE";
    default:                   return "?";
    }
}

int main(int argc, char **argv) {
    const char *path = argc > 1 ? argv[1] : VGPU_VMEM_PATH;
    vmem_region_t *vm = vgpu_region_attach(
        path, sizeof(vmem_region_t), VGPU_VMEM_MAGIC, false, NULL);
    if (!vm) {
        fprintf(stderr, "no vmem region at %s\n", path);
        return 1;
    }
    printf("vmem_node %s: cap=%u created=%llu\n", path,
           vm->record_cap, (unsigned long long)vm->created_ns);
    for (int d = 0; d < MAX_DEVICE_COUNT; d++) {
        unsigned long long used = (unsigned long long)__atomic_load_n(
            &vm->counters[d].vmem_used, __ATOMIC_ACQUIRE);
        unsigned long long hooked = (unsigned long long)__atomic_load_n(
            &vm->counters[d].dev_hooked_used, __ATOMIC_ACQUIRE);
        if (used || hooked)
            printf("dev %d: hooked=%llu MiB vmem=%llu MiB\n", d,
                   hooked >> 20, used >> 20);
    }
    int live = 0;
    for (uint32_t i = 0; i < vm->record_cap && i < MAX_VMEM_RECORDS;
         i++) {
        vmem_record_t *r = &vm->records[i];
        uint64_t size = __atomic_load_n(&r->size, __ATOMIC_ACQUIRE);
        if (!size) continue;
        live++;
        printf("  rec %u: dev=%d pid=%d kind=%s dptr=%#llx "
               "size=%llu MiB\n",
               i, r->device, r->pid, kind_name(r->kind),
               (unsigned long long)r->dptr,
               (unsigned long long)(size >> 20));
    }
    printf("%d live records\n", live);
    return 0;
}
