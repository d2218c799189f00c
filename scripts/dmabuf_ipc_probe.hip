// Probe: can a hipMalloc allocation cross processes as a dmabuf fd
// (hipMemGetHandleForAddressRange -> SCM_RIGHTS -> hipImportExternalMemory)
// — including sizes >= 2 GiB where hipIpcOpenMemHandle hangs
// (scripts/ipc_size_probe.py)? If yes, this is the native big-allocation
// local path for the store.
//
//   hipcc --offload-arch=gfx950 -o dmabuf_ipc_probe scripts/dmabuf_ipc_probe.hip
//   ./dmabuf_ipc_probe 1 3   # sizes in GiB
#include <hip/hip_runtime.h>
#include <sys/socket.h>
#include <sys/wait.h>
#include <unistd.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>

#define CK(x)                                                       \
    do {                                                            \
        hipError_t e_ = (x);                                        \
        if (e_ != hipSuccess) {                                     \
            fprintf(stderr, "  %s -> %s\n", #x, hipGetErrorString(e_)); \
            return 1;                                               \
        }                                                           \
    } while (0)

static int send_fd(int sock, int fd, uint64_t sz, uint64_t off) {
    struct msghdr msg = {};
    char buf[CMSG_SPACE(sizeof(int))] = {};
    uint64_t payload[2] = {sz, off};
    struct iovec io = {payload, sizeof(payload)};
    msg.msg_iov = &io;
    msg.msg_iovlen = 1;
    msg.msg_control = buf;
    msg.msg_controllen = sizeof(buf);
    struct cmsghdr* c = CMSG_FIRSTHDR(&msg);
    c->cmsg_level = SOL_SOCKET;
    c->cmsg_type = SCM_RIGHTS;
    c->cmsg_len = CMSG_LEN(sizeof(int));
    memcpy(CMSG_DATA(c), &fd, sizeof(int));
    return sendmsg(sock, &msg, 0) < 0 ? -1 : 0;
}

static int recv_fd(int sock, int* fd, uint64_t* sz, uint64_t* off) {
    struct msghdr msg = {};
    char buf[CMSG_SPACE(sizeof(int))] = {};
    uint64_t payload[2];
    struct iovec io = {payload, sizeof(payload)};
    msg.msg_iov = &io;
    msg.msg_iovlen = 1;
    msg.msg_control = buf;
    msg.msg_controllen = sizeof(buf);
    if (recvmsg(sock, &msg, 0) <= 0) return -1;
    struct cmsghdr* c = CMSG_FIRSTHDR(&msg);
    if (!c || c->cmsg_type != SCM_RIGHTS) return -1;
    memcpy(fd, CMSG_DATA(c), sizeof(int));
    *sz = payload[0];
    *off = payload[1];
    return 0;
}

static int child_main(int sock) {
    CK(hipSetDevice(0));
    int fd = -1;
    uint64_t sz = 0, off = 0;
    if (recv_fd(sock, &fd, &sz, &off) != 0) {
        fprintf(stderr, "  child recv_fd failed\n");
        return 1;
    }
    hipExternalMemoryHandleDesc hd = {};
    hd.type = hipExternalMemoryHandleTypeOpaqueFd;
    hd.handle.fd = fd;
    hd.size = sz + off;
    hipExternalMemory_t ext = nullptr;
    CK(hipImportExternalMemory(&ext, &hd));
    hipExternalMemoryBufferDesc bd = {};
    bd.offset = off;
    bd.size = sz;
    void* ptr = nullptr;
    CK(hipExternalMemoryGetMappedBuffer(&ptr, ext, &bd));
    // verify first + last 8 bytes (parent wrote a pattern)
    uint64_t head = 0, tail = 0;
    CK(hipMemcpy(&head, ptr, 8, hipMemcpyDeviceToHost));
    CK(hipMemcpy(&tail, (char*)ptr + sz - 8, 8, hipMemcpyDeviceToHost));
    printf("  child: mapped %p head=%llx tail=%llx -> %s\n", ptr,
           (unsigned long long)head, (unsigned long long)tail,
           (head == 0x1122334455667788ull && tail == 0x99aabbccddeeff00ull)
               ? "OK"
               : "MISMATCH");
    hipDestroyExternalMemory(ext);
    return 0;
}

static int run(double gib) {
    size_t sz = (size_t)(gib * (1ull << 30));
    sz &= ~size_t(7);
    printf("size %.2f GiB:\n", gib);
    int socks[2];
    if (socketpair(AF_UNIX, SOCK_STREAM, 0, socks) != 0) return 1;
    pid_t pid = fork();
    if (pid == 0) {
        close(socks[0]);
        _exit(child_main(socks[1]));
    }
    close(socks[1]);
    CK(hipSetDevice(0));
    void* p = nullptr;
    CK(hipMalloc(&p, sz));
    uint64_t head = 0x1122334455667788ull, tail = 0x99aabbccddeeff00ull;
    CK(hipMemcpy(p, &head, 8, hipMemcpyHostToDevice));
    CK(hipMemcpy((char*)p + sz - 8, &tail, 8, hipMemcpyHostToDevice));
    int fd = -1;
    uint64_t off = 0;
    hipError_t e = hipMemGetHandleForAddressRange(
        &fd, (hipDeviceptr_t)p, sz, hipMemRangeHandleTypeDmaBufFd, 0);
    if (e != hipSuccess) {
        printf("  export failed: %s\n", hipGetErrorString(e));
        kill(pid, SIGKILL);
        return 1;
    }
    printf("  parent: exported dmabuf fd=%d\n", fd);
    if (send_fd(socks[0], fd, sz, off) != 0) return 1;
    int status = 0;
    for (int i = 0; i < 300; i++) {
        if (waitpid(pid, &status, WNOHANG) == pid) {
            printf("  child exit=%d\n", WEXITSTATUS(status));
            hipFree(p);
            return 0;
        }
        usleep(100000);
    }
    printf("  CHILD HANGS (>30s)\n");
    kill(pid, SIGKILL);
    return 1;
}

int main(int argc, char** argv) {
    if (argc < 2) {
        run(1.0);
        run(3.0);
        return 0;
    }
    for (int i = 1; i < argc; i++) run(atof(argv[i]));
    return 0;
}
