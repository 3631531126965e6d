/* device_client.c — the tiny registration CLI the shim fork/execs in
 * client mode (reference cmd/device-client rebuilt in C so containers
 * need no Go/Python runtime).  Speaks the registry's JSON-line unix
 * protocol; exit 0 on {"ok": true}.                                   */
#define _GNU_SOURCE
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

int main(int argc, char **argv) {
    const char *sock_path = "/etc/vgpu-manager/registry/socket.sock";
    const char *pod_uid = NULL, *container = NULL;
    long pid = 0;
    for (int i = 1; i + 1 < argc; i += 2) {
        if (strcmp(argv[i], "--socket") == 0) sock_path = argv[i + 1];
        else if (strcmp(argv[i], "--pod-uid") == 0) pod_uid = argv[i + 1];
        else if (strcmp(argv[i], "--container") == 0)
            container = argv[i + 1];
        else if (strcmp(argv[i], "--pid") == 0)
            pid = atol(argv[i + 1]);
    }
    if (!pod_uid || !container) {
        fprintf(stderr,
                "usage: device-client --pod-uid U --container C "
                "[--socket S] [--pid P]\n");
        return 2;
    }
    int fd = socket(AF_UNIX, SOCK_STREAM, 0);
    if (fd < 0) return 1;
    struct sockaddr_un addr = {0};
    addr.sun_family = AF_UNIX;
    snprintf(addr.sun_path, sizeof(addr.sun_path), "%s", sock_path);
    if (connect(fd, (struct sockaddr *)&addr, sizeof(addr)) != 0) {
        perror("connect");
        return 1;
    }
    char req[512];
    if (pid > 0)
        snprintf(req, sizeof(req),
                 "{\"pod_uid\":\"%s\",\"container_name\":\"%s\","
                 "\"pids\":[%ld,%d]}\n",
                 pod_uid, container, pid, (int)getpid());
    else
        snprintf(req, sizeof(req),
                 "{\"pod_uid\":\"%s\",\"container_name\":\"%s\","
                 "\"pids\":[%d]}\n",
                 pod_uid, container, (int)getpid());
    if (write(fd, req, strlen(req)) < 0) return 1;
    char resp[1024] = {0};
    ssize_t n = read(fd, resp, sizeof(resp) - 1);
    close(fd);
    if (n <= 0) return 1;
    return strstr(resp, "\"ok\": true") || strstr(resp, "\"ok\":true")
               ? 0
               : 1;
}
