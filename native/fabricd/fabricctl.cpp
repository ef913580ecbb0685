// fabricctl — control CLI for fabricd (the `nvidia-imex-ctl` analog).
//
//   fabricctl -q [-p port] [-H host]   query status; prints READY/NOT_READY,
//                                      exit 0 iff READY
//   fabricctl peers / probe            diagnostics

#include <arpa/inet.h>
#include <netdb.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/socket.h>
#include <unistd.h>

#include <string>

static int connect_to(const char* host, int port) {
    addrinfo hints{}, *res = nullptr;
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    char portstr[16];
    snprintf(portstr, sizeof portstr, "%d", port);
    if (getaddrinfo(host, portstr, &hints, &res) != 0 || !res) return -1;
    int fd = socket(res->ai_family, res->ai_socktype, res->ai_protocol);
    if (fd >= 0) {
        timeval tv{5, 0};
        setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof tv);
        if (connect(fd, res->ai_addr, res->ai_addrlen) != 0) {
            close(fd);
            fd = -1;
        }
    }
    freeaddrinfo(res);
    return fd;
}

int main(int argc, char** argv) {
    const char* host = "127.0.0.1";
    int port = 50005;
    std::string cmd = "STATUS";
    for (int i = 1; i < argc; ++i) {
        if (strcmp(argv[i], "-q") == 0) cmd = "STATUS";
        else if (strcmp(argv[i], "peers") == 0) cmd = "PEERS";
        else if (strcmp(argv[i], "probe") == 0) cmd = "PROBE";
        else if (strcmp(argv[i], "metrics") == 0) cmd = "METRICS";
        else if (strcmp(argv[i], "burn") == 0) cmd = "BURN";
        else if (strcmp(argv[i], "-p") == 0 && i + 1 < argc) port = atoi(argv[++i]);
        else if (strcmp(argv[i], "-H") == 0 && i + 1 < argc) host = argv[++i];
    }
    int fd = connect_to(host, port);
    if (fd < 0) {
        printf("NOT_READY no-daemon\n");
        return 1;
    }
    std::string msg = cmd + "\n";
    send(fd, msg.data(), msg.size(), 0);
    char buf[4096];
    ssize_t n;
    std::string out;
    while ((n = recv(fd, buf, sizeof buf, 0)) > 0) out.append(buf, n);
    close(fd);
    fputs(out.c_str(), stdout);
    return out.rfind("READY", 0) == 0 ? 0 : 1;
}
