#include "http_server.h"

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>
#include <vector>

namespace mi355x {

HttpServer::HttpServer(std::string bind_addr, int port, Handler metrics, ReadyFn ready)
    : bind_addr_(std::move(bind_addr)), port_(port), metrics_(std::move(metrics)),
      ready_(std::move(ready))
{
}

HttpServer::~HttpServer() { stop(); }

bool HttpServer::start(std::string* err)
{
    int fd = ::socket(AF_INET, SOCK_STREAM, 0);
    listen_fd_ = fd;
    if (fd < 0) {
        if (err) *err = "socket() failed";
        return false;
    }
    int one = 1;
    setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr;
    std::memset(&addr, 0, sizeof(addr));
    addr.sin_family = AF_INET;
    addr.sin_port = htons((uint16_t)port_);
    if (bind_addr_.empty() || bind_addr_ == "0.0.0.0")
        addr.sin_addr.s_addr = INADDR_ANY;
    else if (inet_pton(AF_INET, bind_addr_.c_str(), &addr.sin_addr) != 1) {
        if (err) *err = "bad bind address " + bind_addr_;
        ::close(fd);
        listen_fd_ = -1;
        return false;
    }
    if (::bind(fd, (sockaddr*)&addr, sizeof(addr)) != 0) {
        if (err) *err = "bind failed on port " + std::to_string(port_);
        ::close(fd);
        listen_fd_ = -1;
        return false;
    }
    if (port_ == 0) {
        socklen_t len = sizeof(addr);
        getsockname(fd, (sockaddr*)&addr, &len);
        port_ = ntohs(addr.sin_port);
    }
    if (::listen(fd, 64) != 0) {
        if (err) *err = "listen failed";
        ::close(fd);
        listen_fd_ = -1;
        return false;
    }
    stop_ = false;
    thread_ = std::thread([this] { accept_loop(); });
    return true;
}

void HttpServer::stop()
{
    stop_ = true;
    int fd = listen_fd_.exchange(-1);
    if (fd >= 0) {
        ::shutdown(fd, SHUT_RDWR);
        // close AFTER joining the accept loop so its poll/accept never
        // touches a recycled fd number
        if (thread_.joinable()) thread_.join();
        ::close(fd);
        return;
    }
    if (thread_.joinable()) thread_.join();
}

void HttpServer::accept_loop()
{
    while (!stop_) {
        int lfd = listen_fd_.load();
        if (lfd < 0) break;
        pollfd p{lfd, POLLIN, 0};
        int rc = ::poll(&p, 1, 250);
        if (stop_) break;
        if (rc <= 0) continue;
        int fd = ::accept(lfd, nullptr, nullptr);
        if (fd < 0) continue;
        // handle inline: requests are tiny and render is fast; a stuck
        // client can't stall us thanks to the send timeout.
        timeval tv{5, 0};
        setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
        setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
        handle(fd);
        ::close(fd);
    }
}

static void send_all(int fd, const std::string& s)
{
    size_t off = 0;
    while (off < s.size()) {
        ssize_t n = ::send(fd, s.data() + off, s.size() - off, MSG_NOSIGNAL);
        if (n <= 0) return;
        off += (size_t)n;
    }
}

void HttpServer::handle(int fd)
{
    char buf[4096];
    ssize_t n = ::recv(fd, buf, sizeof(buf) - 1, 0);
    if (n <= 0) return;
    buf[n] = 0;
    std::string req(buf);
    auto line_end = req.find("\r\n");
    std::string line = line_end == std::string::npos ? req : req.substr(0, line_end);

    std::string status = "200 OK", body, ctype = "text/plain; charset=utf-8";
    if (line.rfind("GET /metrics", 0) == 0) {
        body = metrics_();
        ctype = "text/plain; version=0.0.4; charset=utf-8";
    } else if (line.rfind("GET /healthz", 0) == 0) {
        body = "ok\n";
    } else if (line.rfind("GET /readyz", 0) == 0) {
        if (ready_ && ready_()) {
            body = "ready\n";
        } else {
            status = "503 Service Unavailable";
            body = "no successful GPU sample yet\n";
        }
    } else if (line.rfind("GET /", 0) == 0) {
        status = "404 Not Found";
        body = "see /metrics\n";
    } else {
        status = "405 Method Not Allowed";
        body = "GET only\n";
    }

    std::string resp = "HTTP/1.1 " + status + "\r\nContent-Type: " + ctype +
                       "\r\nContent-Length: " + std::to_string(body.size()) +
                       "\r\nConnection: close\r\n\r\n" + body;
    send_all(fd, resp);
}

} // namespace mi355x
