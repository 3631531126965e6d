/* util.h — env parsing, tunables, container PID sets. */
#ifndef VGPU_UTIL_H
#define VGPU_UTIL_H

#include "hook.h"
#include <stdbool.h>

#ifdef __cplusplus
extern "C" {
#endif

/* getenv with PID-1-environ fallback: an exec'd shell may have dropped
 * the kubelet-injected env; PID 1 of the container always has it.     */
const char *vgpu_getenv(const char *name, char *buf, size_t buflen);

/* parse IEC sizes: "1g" "512mi" "1073741824" -> bytes; -1 on error    */
long long vgpu_parse_size(const char *s);

/* parsed-once env tunables (see hook.h dynamic_config_t).
 * Env surface (HIP_* family, structure-parity with the reference's
 * CUDA_* family, util.c:401-583 there):
 *   VGPU_CU_CONTROLLER        delta|aimd|auto      (default auto)
 *   VGPU_CU_USAGE_THRESHOLD   int %                (default 10)
 *   VGPU_CU_AIMD_MD_DIVISOR   int                  (default 3)
 *   VGPU_CU_AIMD_EFF_RATIO    "7/8"
 *   VGPU_CU_AIMD_AI_BASE_DIV  int                  (default 64)
 *   VGPU_CU_AIMD_DEADBAND_PERMILLE int             (default 20)
 *   VGPU_CU_AIMD_MD_COOLDOWN_CYCLES int            (default 5)
 *   VGPU_CU_AUTO_DEBOUNCE_CYCLES int               (default 3)
 *   VGPU_CU_AUTO_EXTERNAL_UTIL_THRESHOLD permille  (default 50)
 *   VGPU_CU_DELTA_RAMP_FLOOR_DIVISOR int           (default 10)
 *   VGPU_CU_SHARED_BUCKET     0|1                  (default 1)
 *   VGPU_MEM_OVERSOLD         0|1                  (default 0)
 *   VGPU_MEM_ACCOUNT_MODE     ledger|smi|max       (default max)
 *   VGPU_MEM_UVA_ADVISE       0|1                  (default 1)
 *   VGPU_GAP_DISABLE          0|1                  (default 0)
 *   VGPU_LOGGER_LEVEL         0-5
 */
const dynamic_config_t *vgpu_dynconfig(void);

/* memory account modes */
#define MEM_ACCOUNT_LEDGER 0
#define MEM_ACCOUNT_SMI    1
#define MEM_ACCOUNT_MAX    2

/* container PID set: loaded from pids.config if present, else built
 * from this process tree.  Sorted; membership by binary search.
 *
 * `host_pids` mirrors `pids` translated to the HOST pid namespace:
 * KFD sysfs and amd-smi report host pids, so a namespaced container
 * can only attribute per-process data after translation.  The bridge
 * is the PASID: our own /proc/<pid>/fdinfo carries "pasid: N" for
 * amdgpu fds, and /sys/class/kfd/kfd/proc/<hostpid>/pasid is global —
 * matching them maps ns pid -> host pid (the reference reaches host
 * pids via its mounted .host_proc / peercred registry instead).       */
typedef struct {
    int32_t pids[MAX_DEVICE_PIDS];
    int32_t host_pids[MAX_DEVICE_PIDS]; /* sorted; 0 = unresolved     */
    int     count;
    int     host_count;
    int32_t self_host_pid;  /* VRAM-probe result (smi_self_host_pid)  */
    int     host_native;    /* pids came from the host side already    */
    uint64_t loaded_ns;
} pid_set_t;

/* (re)load the container pid set; returns number of pids (>=0).       */
int  vgpu_load_pid_set(pid_set_t *set);
bool vgpu_pid_set_contains(const pid_set_t *set, int32_t pid);
/* translate ns pid -> host pid via the KFD pasid bridge; returns the
 * input when no translation is needed or possible.                    */
int32_t vgpu_pid_to_host(int32_t ns_pid);
/* refresh host_pids from pids (cheap when already fully resolved).    */
void vgpu_pid_set_resolve_host(pid_set_t *set);
/* true when the set already has a real ns->host mapping (translated
 * entry or a probed self_host_pid) — i.e. no further probe needed.    */
bool vgpu_pid_set_translated(const pid_set_t *set);
/* instantaneous CUs occupied by the set's host pids (KFD sysfs).      */
uint32_t vgpu_kfd_cu_occupancy_sum(const pid_set_t *set);
/* ours + other tenants' instantaneous CU occupancy.                   */
void vgpu_kfd_cu_occupancy2(const pid_set_t *set, uint32_t *ours,
                            uint32_t *others);

#ifdef __cplusplus
}
#endif
#endif
