// http_server.h — minimal HTTP/1.1 server for the exporter endpoints.
//
// Serves what the reference exporter serves on :9400 (dcgm-exporter.yaml:
// 31-32,39-41) plus the liveness/readiness endpoints the reference lacks
// (SURVEY.md §5.3 flags the missing probes):
//   GET /metrics  -> Prometheus text format (render callback)
//   GET /healthz  -> 200 once the process is up (liveness)
//   GET /readyz   -> 200 after the first successful counter sample
//                    (readiness), 503 before
// Single accept thread + short-lived handler threads; Prometheus scrapes at
// 1 s (kube-prometheus-stack-values.yaml:5) are trivially sustained.

#pragma once

#include <atomic>
#include <functional>
#include <string>
#include <thread>

namespace mi355x {

class HttpServer {
  public:
    using Handler = std::function<std::string()>;   // returns /metrics body
    using ReadyFn = std::function<bool()>;

    HttpServer(std::string bind_addr, int port, Handler metrics, ReadyFn ready);
    ~HttpServer();

    // returns false + err on bind failure. port 0 picks an ephemeral port
    // (tests); bound_port() reports it.
    bool start(std::string* err);
    void stop();
    int bound_port() const { return port_; }

  private:
    void accept_loop();
    void handle(int fd);

    std::string bind_addr_;
    int port_;
    Handler metrics_;
    ReadyFn ready_;
    std::atomic<int> listen_fd_{-1};
    std::atomic<bool> stop_{false};
    std::thread thread_;
};

} // namespace mi355x
