/* mpix — bootstrap implementations: MPI allgather or a tiny TCP store.
 *
 * The TCP store protocol (env mode): rank 0 listens on
 * MASTER_ADDR:MASTER_PORT + MPIX_PORT_OFFSET (default 31; offset avoids the
 * torchrun c10d store living on MASTER_PORT itself).  Every other rank
 * connects once and keeps the socket.  Each collective round:
 *   client -> server : [u32 rank][u32 nbytes][blob]
 *   server -> client : [all blobs in rank order]
 * A barrier is an allgather of 1 byte.
 */
#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cerrno>
#include <chrono>
#include <cstring>
#include <string>
#include <thread>
#include <vector>

#include <mpi.h>

#include "bootstrap.h"
#include "../internal.h"

namespace mpix {

/* ----------------------------------------------------------------- MPI mode */

class MpiBootstrap : public Bootstrap {
public:
    int allgather(const void *mine, void *all, size_t blob_size) override {
        return MPI_Allgather(mine, (int)blob_size, MPI_BYTE, all,
                             (int)blob_size, MPI_BYTE, MPI_COMM_WORLD);
    }
    int barrier() override { return MPI_Barrier(MPI_COMM_WORLD); }
};

/* ----------------------------------------------------------------- env mode */

static int read_full(int fd, void *buf, size_t n)
{
    char *p = (char *)buf;
    while (n > 0) {
        ssize_t r = read(fd, p, n);
        if (r < 0) {
            if (errno == EINTR) continue;
            return -1;
        }
        if (r == 0) return -1;
        p += r; n -= (size_t)r;
    }
    return 0;
}

static int write_full(int fd, const void *buf, size_t n)
{
    const char *p = (const char *)buf;
    while (n > 0) {
        ssize_t r = write(fd, p, n);
        if (r < 0) {
            if (errno == EINTR) continue;
            return -1;
        }
        p += r; n -= (size_t)r;
    }
    return 0;
}

class TcpBootstrap : public Bootstrap {
public:
    TcpBootstrap(int rank, int size) : rank_(rank), size_(size) {}

    int init() {
        const char *addr = getenv("MASTER_ADDR");
        if (!addr || !*addr) addr = "127.0.0.1";
        int base_port = 29500;
        if (const char *p = getenv("MASTER_PORT")) base_port = atoi(p);
        int off = 31;
        if (const char *p = getenv("MPIX_PORT_OFFSET")) off = atoi(p);
        port_ = base_port + off;
        if (size_ == 1) return 0;
        if (rank_ == 0) return init_server();
        return init_client(addr);
    }

    ~TcpBootstrap() override {
        for (int fd : peer_fds_) if (fd >= 0) close(fd);
        if (sock_ >= 0) close(sock_);
        if (listen_fd_ >= 0) close(listen_fd_);
    }

    int allgather(const void *mine, void *all, size_t blob_size) override {
        if (size_ == 1) {
            memcpy(all, mine, blob_size);
            return 0;
        }
        if (rank_ == 0) {
            memcpy((char *)all, mine, blob_size);
            /* collect one blob from every client */
            for (int i = 1; i < size_; i++) {
                uint32_t r = 0, nb = 0;
                int fd = peer_fds_[i];
                if (read_full(fd, &r, 4) || read_full(fd, &nb, 4)) return -1;
                if (nb != blob_size || r == 0 || (int)r >= size_) return -1;
                if (read_full(fd, (char *)all + (size_t)r * blob_size, blob_size))
                    return -1;
            }
            for (int i = 1; i < size_; i++)
                if (write_full(peer_fds_[i], all, blob_size * (size_t)size_))
                    return -1;
            return 0;
        }
        uint32_t r = (uint32_t)rank_, nb = (uint32_t)blob_size;
        if (write_full(sock_, &r, 4) || write_full(sock_, &nb, 4) ||
            write_full(sock_, mine, blob_size))
            return -1;
        return read_full(sock_, all, blob_size * (size_t)size_);
    }

    int barrier() override {
        char b = 0;
        std::vector<char> all((size_t)size_);
        return allgather(&b, all.data(), 1);
    }

private:
    int init_server() {
        listen_fd_ = socket(AF_INET, SOCK_STREAM, 0);
        if (listen_fd_ < 0) return -1;
        int one = 1;
        setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
        sockaddr_in sa{};
        sa.sin_family = AF_INET;
        sa.sin_addr.s_addr = htonl(INADDR_ANY);
        sa.sin_port = htons((uint16_t)port_);
        if (bind(listen_fd_, (sockaddr *)&sa, sizeof(sa)) < 0) {
            MPIX_ERR("bootstrap bind :%d failed: %s", port_, strerror(errno));
            return -1;
        }
        if (listen(listen_fd_, size_) < 0) return -1;
        peer_fds_.assign(size_, -1);
        for (int i = 1; i < size_; i++) {
            int fd = accept(listen_fd_, nullptr, nullptr);
            if (fd < 0) return -1;
            int on = 1;
            setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &on, sizeof(on));
            uint32_t r = 0;
            if (read_full(fd, &r, 4) || r == 0 || (int)r >= size_) {
                close(fd);
                return -1;
            }
            peer_fds_[r] = fd;
        }
        for (int i = 1; i < size_; i++) if (peer_fds_[i] < 0) return -1;
        return 0;
    }

    int init_client(const char *addr) {
        addrinfo hints{}, *res = nullptr;
        hints.ai_family = AF_INET;
        hints.ai_socktype = SOCK_STREAM;
        char portstr[16];
        snprintf(portstr, sizeof(portstr), "%d", port_);
        if (getaddrinfo(addr, portstr, &hints, &res) != 0 || !res) return -1;
        /* rank 0 may not be listening yet: retry for up to ~60 s */
        int fd = -1;
        for (int attempt = 0; attempt < 1200; attempt++) {
            fd = socket(AF_INET, SOCK_STREAM, 0);
            if (fd < 0) break;
            if (connect(fd, res->ai_addr, res->ai_addrlen) == 0) break;
            close(fd);
            fd = -1;
            std::this_thread::sleep_for(std::chrono::milliseconds(50));
        }
        freeaddrinfo(res);
        if (fd < 0) {
            MPIX_ERR("bootstrap connect to %s:%d failed", addr, port_);
            return -1;
        }
        int on = 1;
        setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &on, sizeof(on));
        uint32_t r = (uint32_t)rank_;
        if (write_full(fd, &r, 4)) { close(fd); return -1; }
        sock_ = fd;
        return 0;
    }

    int rank_, size_, port_ = 0;
    int listen_fd_ = -1;
    int sock_ = -1;                /* client: connection to rank 0 */
    std::vector<int> peer_fds_;    /* server: connections from clients */
};

Bootstrap *make_bootstrap(int rank, int size, bool mpi_mode)
{
    if (mpi_mode) return new MpiBootstrap();
    auto *b = new TcpBootstrap(rank, size);
    if (b->init() != 0) {
        delete b;
        return nullptr;
    }
    return b;
}

} /* namespace mpix */
