/* abi_probe.c — prints offsetof/sizeof of every shared ABI struct as
 * "name=value" lines.  The Python ctypes mirrors (vgpu_manager_amd/config)
 * are asserted against this output in tests/test_abi_layout.py, pinning
 * the cross-language ABI from both sides (the reference pins Go vs C the
 * same way: pkg/config/vgpu/vgpu_config_test.go + hook.h _Static_asserts).
 */
#include "../include/hook.h"
#include <stdio.h>

#define P(expr) printf(#expr "=%zu\n", (size_t)(expr))

int main(void) {
    P(sizeof(region_header_t));
    P(sizeof(device_t));
    P(offsetof(device_t, seq));
    P(offsetof(device_t, flags));
    P(offsetof(device_t, total_memory));
    P(offsetof(device_t, core_limit));
    P(offsetof(device_t, soft_core_limit));
    P(offsetof(device_t, host_index));
    P(offsetof(device_t, uuid));
    P(sizeof(resource_data_t));
    P(offsetof(resource_data_t, pod_uid));
    P(offsetof(resource_data_t, pod_name));
    P(offsetof(resource_data_t, pod_namespace));
    P(offsetof(resource_data_t, container_name));
    P(offsetof(resource_data_t, device_count));
    P(offsetof(resource_data_t, compute_policy));
    P(offsetof(resource_data_t, oversold));
    P(offsetof(resource_data_t, devices));
    P(sizeof(pids_data_t));
    P(offsetof(pids_data_t, pid_count));
    P(offsetof(pids_data_t, updated_ns));
    P(offsetof(pids_data_t, pids));
    P(sizeof(util_proc_t));
    P(sizeof(device_util_t));
    P(offsetof(device_util_t, seq));
    P(offsetof(device_util_t, dev_busy_permille));
    P(offsetof(device_util_t, sample_ns));
    P(offsetof(device_util_t, proc_count));
    P(offsetof(device_util_t, vram_used_bytes));
    P(offsetof(device_util_t, procs));
    P(sizeof(util_region_t));
    P(offsetof(util_region_t, device_count));
    P(offsetof(util_region_t, heartbeat_ns));
    P(offsetof(util_region_t, devices));
    P(sizeof(vmem_record_t));
    P(offsetof(vmem_record_t, state));
    P(offsetof(vmem_record_t, kind));
    P(offsetof(vmem_record_t, dptr));
    P(offsetof(vmem_record_t, size));
    P(offsetof(vmem_record_t, pid));
    P(offsetof(vmem_record_t, device));
    P(offsetof(vmem_record_t, created_ns));
    P(sizeof(vmem_dev_counter_t));
    P(offsetof(vmem_dev_counter_t, vmem_used));
    P(offsetof(vmem_dev_counter_t, dev_hooked_used));
    P(sizeof(vmem_region_t));
    P(offsetof(vmem_region_t, record_cap));
    P(offsetof(vmem_region_t, created_ns));
    P(offsetof(vmem_region_t, counters));
    P(offsetof(vmem_region_t, records));
    P(sizeof(sm_node_dev_t));
    P(offsetof(sm_node_dev_t, tokens));
    P(offsetof(sm_node_dev_t, pool_size));
    P(offsetof(sm_node_dev_t, refill_owner_pid));
    P(offsetof(sm_node_dev_t, controller_kind));
    P(offsetof(sm_node_dev_t, refill_ns));
    P(offsetof(sm_node_dev_t, cur_share));
    P(offsetof(sm_node_dev_t, sample_seq));
    P(offsetof(sm_node_dev_t, util_permille));
    P(offsetof(sm_node_dev_t, dev_busy_permille));
    P(offsetof(sm_node_dev_t, sample_ns));
    P(sizeof(sm_node_region_t));
    P(offsetof(sm_node_region_t, device_count));
    P(offsetof(sm_node_region_t, created_ns));
    P(offsetof(sm_node_region_t, devices));
    return 0;
}
