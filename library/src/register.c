/* register.c — client-registration mode.
 *
 * When the node agent runs the registry server (DevicePluginClientMode),
 * the container cannot see the host /proc; instead each process
 * registers its PID by fork/exec'ing the mounted `device-client` CLI
 * against the registry unix socket.  The server resolves the calling
 * container via SO_PEERCRED + cgroup walk and persists pids.config.
 * (Reference: library/src/register.c + pkg/device/registry.)
 */
#define _GNU_SOURCE
#include "state.h"
#include "shm.h"

#include <stdio.h>
#include <stdlib.h>
#include <sys/wait.h>
#include <unistd.h>

void vgpu_register_client(void) {
    char sbuf[256], cbuf[256];
    const char *sock = vgpu_getenv("VGPU_REGISTRY_SOCKET_OVERRIDE",
                                   sbuf, sizeof(sbuf));
    if (!sock) sock = VGPU_REGISTRY_SOCKET;
    const char *cli = vgpu_getenv("VGPU_DEVICE_CLIENT_OVERRIDE", cbuf,
                                  sizeof(cbuf));
    if (!cli) cli = VGPU_DEVICE_CLIENT;
    if (access(sock, F_OK) != 0 || access(cli, X_OK) != 0)
        return; /* not client mode */
    char buf[128];
    const char *uid = vgpu_getenv("VGPU_POD_UID", buf, sizeof(buf));
    char buf2[128];
    const char *cont = vgpu_getenv("VGPU_CONTAINER_NAME", buf2, sizeof(buf2));
    if (!uid || !cont) return;

    pid_t pid = fork();
    if (pid < 0) return;
    if (pid == 0) {
        char pidstr[16];
        snprintf(pidstr, sizeof(pidstr), "%d", (int)getppid());
        execl(cli, "device-client", "--socket",
              sock, "--pod-uid", uid, "--container", cont,
              "--pid", pidstr, (char *)NULL);
        _exit(127);
    }
    int st = 0;
    waitpid(pid, &st, 0);
    if (!WIFEXITED(st) || WEXITSTATUS(st) != 0)
        LOGGER(LOG_WARN, "device-client registration failed (status %d)", st);
    else
        LOGGER(LOG_DEBUG, "registered pid %d with registry", (int)getpid());
}
