// Native KvState query server: the MI355X-stack equivalent of the
// reference's Netty KvStateServer (one per TaskManager; reached through
// QueryableStateClient, flink-queryable-client/.../QueryClientHelper.java:
// 92,121).  A C++ HTTP/1.1 keep-alive server with its OWN keyed store
// (payload text + parsed fp64 vectors behind a shared_mutex), answering
// point lookups and ALS dot-product predictions entirely off the Python
// GIL — the FastAPI app stays the control plane (ingest, checkpoints,
// SGD); this is the data plane for the serving hot path.
//
// Response JSON shapes byte-match the FastAPI endpoints so the existing
// clients/loadgens work unchanged:
//   GET /state/ALS_MODEL/<key>   {"key":..,"value":[..,..]} | 404 detail
//   GET /als/predict?user=&item= ALSPredict.java:74-86 semantics
//   GET /healthz                 {"ok":true}

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <atomic>
#include <cerrno>
#include <chrono>
#include <cstring>
#include <memory>
#include <shared_mutex>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

namespace {

struct Entry {
    std::string payload;
    std::vector<double> vec;
};

std::string json_escape(const std::string& s) {
    std::string o;
    o.reserve(s.size() + 8);
    for (char c : s) {
        if (c == '"' || c == '\\') { o += '\\'; o += c; }
        else if ((unsigned char)c < 0x20) { o += ' '; }
        else o += c;
    }
    return o;
}

// Double formatting for the predict reply: match Python's f"{pred:f}"
std::string fmt_f(double v) {
    char buf[64];
    snprintf(buf, sizeof(buf), "%f", v);
    return buf;
}

class KvServer {
public:
    KvServer() = default;
    ~KvServer() { stop(); }

    int start(int port) {
        listen_fd_ = socket(AF_INET, SOCK_STREAM, 0);
        int one = 1;
        setsockopt(listen_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
        sockaddr_in addr{};
        addr.sin_family = AF_INET;
        addr.sin_addr.s_addr = htonl(INADDR_LOOPBACK);
        addr.sin_port = htons((uint16_t)port);
        if (bind(listen_fd_, (sockaddr*)&addr, sizeof(addr)) != 0) {
            close(listen_fd_);
            throw std::runtime_error("kvserver bind failed");
        }
        socklen_t alen = sizeof(addr);
        getsockname(listen_fd_, (sockaddr*)&addr, &alen);
        port_ = ntohs(addr.sin_port);
        listen(listen_fd_, 512);
        running_ = true;
        acceptor_ = std::thread([this] { accept_loop(); });
        return port_;
    }

    void stop() {
        if (!running_.exchange(false)) return;
        shutdown(listen_fd_, SHUT_RDWR);
        close(listen_fd_);
        if (acceptor_.joinable()) acceptor_.join();
        // connection threads poll running_ via a 200 ms recv timeout;
        // wait (bounded) until they all exit so the store cannot be
        // destroyed under a live reader
        for (int i = 0; i < 25 && conns_.load() > 0; ++i)
            std::this_thread::sleep_for(std::chrono::milliseconds(100));
    }

    // rows: "<id>,<U|I>,<f;f;...>" — parsed once here, then both the
    // payload echo and the prediction dot run lock-free per request
    // (shared lock) off the GIL.
    void put_rows(const std::vector<std::string>& rows) {
        std::vector<std::pair<std::string, Entry>> parsed;
        parsed.reserve(rows.size());
        {
            py::gil_scoped_release rel;
            for (const auto& row : rows) {
                auto c1 = row.find(',');
                if (c1 == std::string::npos) continue;
                auto c2 = row.find(',', c1 + 1);
                if (c2 == std::string::npos) continue;
                Entry e;
                e.payload = row.substr(c2 + 1);
                const char* p = e.payload.c_str();
                char* q;
                while (*p) {
                    double v = strtod(p, &q);
                    if (q == p) break;
                    e.vec.push_back(v);
                    p = (*q == ';') ? q + 1 : q;
                    if (q == p && *q) break;
                }
                std::string key = row.substr(0, c1) + "-" +
                                  row.substr(c1 + 1, c2 - c1 - 1);
                parsed.emplace_back(std::move(key), std::move(e));
            }
            std::unique_lock<std::shared_mutex> lk(mu_);
            for (auto& kv : parsed) map_[kv.first] = std::move(kv.second);
        }
    }

    size_t size() {
        std::shared_lock<std::shared_mutex> lk(mu_);
        return map_.size();
    }

    int port() const { return port_; }

private:
    void accept_loop() {
        while (running_) {
            int fd = accept(listen_fd_, nullptr, nullptr);
            if (fd < 0) {
                if (!running_) break;
                continue;
            }
            int one = 1;
            setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
            timeval tv{0, 200000};   // recv timeout: running_ poll period
            setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
            conns_.fetch_add(1);
            std::thread([this, fd] {
                serve_conn(fd);
                conns_.fetch_sub(1);
            }).detach();
        }
    }

    void respond(int fd, int code, const std::string& body) {
        const char* status = code == 200 ? "200 OK" : "404 Not Found";
        char head[160];
        int hn = snprintf(head, sizeof(head),
                          "HTTP/1.1 %s\r\nContent-Type: application/json\r\n"
                          "Content-Length: %zu\r\nConnection: keep-alive\r\n"
                          "\r\n",
                          status, body.size());
        std::string out(head, hn);
        out += body;
        size_t off = 0;
        while (off < out.size()) {
            ssize_t w = send(fd, out.data() + off, out.size() - off,
                             MSG_NOSIGNAL);
            if (w <= 0) return;
            off += (size_t)w;
        }
    }

    bool lookup(const std::string& key, Entry& out) {
        std::shared_lock<std::shared_mutex> lk(mu_);
        auto it = map_.find(key);
        if (it == map_.end()) return false;
        out = it->second;
        return true;
    }

    static std::string url_param(const std::string& qs,
                                 const std::string& name) {
        size_t pos = 0;
        while (pos < qs.size()) {
            size_t amp = qs.find('&', pos);
            if (amp == std::string::npos) amp = qs.size();
            size_t eq = qs.find('=', pos);
            if (eq != std::string::npos && eq < amp &&
                qs.compare(pos, eq - pos, name) == 0)
                return qs.substr(eq + 1, amp - eq - 1);
            pos = amp + 1;
        }
        return "";
    }

    void handle(int fd, const std::string& path) {
        if (path.rfind("/state/ALS_MODEL/", 0) == 0) {
            std::string key = path.substr(17);
            Entry e;
            if (!lookup(key, e)) {
                respond(fd, 404,
                        "{\"detail\":\"unknown key: " + json_escape(key) +
                            "\"}");
                return;
            }
            std::string k = json_escape(key);
            respond(fd, 200,
                    "{\"key\":\"" + k + "\",\"value\":[\"" + k + "\",\"" +
                        json_escape(e.payload) + "\"]}");
            return;
        }
        if (path.rfind("/als/predict?", 0) == 0) {
            std::string qs = path.substr(13);
            std::string user = url_param(qs, "user");
            std::string item = url_param(qs, "item");
            Entry u, v;
            if (!lookup(user + "-U", u) || !lookup(item + "-I", v)) {
                respond(fd, 200,
                        "{\"found\":false,\"message\":\"User or Item "
                        "Factors do not exist in the model for the query: " +
                            json_escape(user) + "," + json_escape(item) +
                            "\"}");
                return;
            }
            double dot = 0.0;
            size_t n = std::min(u.vec.size(), v.vec.size());
            for (size_t i = 0; i < n; ++i) dot += u.vec[i] * v.vec[i];
            char num[32];
            snprintf(num, sizeof(num), "%.17g", dot);
            respond(fd, 200,
                    std::string("{\"found\":true,\"prediction\":") + num +
                        ",\"formatted\":\"ALS Prediction =  " + fmt_f(dot) +
                        " \"}");
            return;
        }
        if (path == "/healthz") {
            respond(fd, 200, "{\"ok\":true}");
            return;
        }
        respond(fd, 404, "{\"detail\":\"not found\"}");
    }

    void serve_conn(int fd) {
        std::string buf;
        char tmp[4096];
        while (running_) {
            // read one request head (hot-path requests have no body)
            size_t hdr_end;
            while ((hdr_end = buf.find("\r\n\r\n")) == std::string::npos) {
                ssize_t r = recv(fd, tmp, sizeof(tmp), 0);
                if (r == 0) { close(fd); return; }
                if (r < 0) {
                    if ((errno == EAGAIN || errno == EWOULDBLOCK ||
                         errno == EINTR) && running_)
                        continue;   // timeout tick: re-check running_
                    close(fd);
                    return;
                }
                buf.append(tmp, (size_t)r);
                if (buf.size() > 1 << 20) { close(fd); return; }
            }
            if (!running_) { close(fd); return; }
            // request line: METHOD SP PATH SP VERSION
            size_t sp1 = buf.find(' ');
            size_t sp2 = buf.find(' ', sp1 + 1);
            if (sp1 == std::string::npos || sp2 == std::string::npos) {
                close(fd);
                return;
            }
            std::string path = buf.substr(sp1 + 1, sp2 - sp1 - 1);
            handle(fd, path);
            buf.erase(0, hdr_end + 4);
        }
        close(fd);
    }

    std::unordered_map<std::string, Entry> map_;
    std::shared_mutex mu_;
    int listen_fd_ = -1;
    int port_ = 0;
    std::atomic<bool> running_{false};
    std::atomic<int> conns_{0};
    std::thread acceptor_;
};

}  // namespace

void register_kvserver(py::module_& m) {
    py::class_<KvServer, std::shared_ptr<KvServer>>(m, "KvServer")
        .def(py::init<>())
        .def("start", &KvServer::start, py::arg("port") = 0,
             "Bind 127.0.0.1:<port> (0 = ephemeral) and serve; returns "
             "the bound port.")
        .def("stop", &KvServer::stop)
        .def("put_rows", &KvServer::put_rows)
        .def("size", &KvServer::size)
        .def_property_readonly("port", &KvServer::port);
}
