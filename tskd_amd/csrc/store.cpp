// tskd store — embedded prediction store + age table (C++17, mmap).
//
// Replaces the reference's MySQL server + mysql-connector
// with a first-party embedded store exposing exactly the four queries the
// reference issues:
//   1. INSERT INTO predictions (SUBJECT_ID, PRED_TIME, RISK_SCORE)
//      (reference predictStream.py:175)
//   2. latest prediction for a patient        (predictStream.py:187)
//   3. predictions since a timestamp ("today's") (plotData.py:198)
//   4. age lookup by subject id               (predictStream.py:30,144-151)
//
// The prediction log is an append-only mmap'd file (same multi-process
// header discipline as the bus: process-shared mutex + atomic committed
// counter), so the inference service and the dashboard are separate OS
// processes sharing one store with no daemon.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <atomic>
#include <cstring>
#include <map>
#include <optional>
#include <stdexcept>
#include <string>
#include <tuple>
#include <vector>

#include <fcntl.h>
#include <pthread.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

namespace py = pybind11;

namespace {

constexpr uint64_t MAGIC = 0x54534b4453544f52ull;  // "TSKDSTOR"
constexpr size_t HDR_SIZE = 4096;
constexpr size_t INITIAL_CAP = 1 << 20;

struct StoreHeader {
    uint64_t magic;
    uint32_t version;
    uint32_t pad_;
    pthread_mutex_t mtx;
    std::atomic<uint64_t> committed;  // record COUNT
};

struct Rec {
    int32_t subject_id;
    int32_t pad_;
    int64_t pred_time_us;
    float risk_score;
    float pad2_;
};
static_assert(sizeof(Rec) == 24, "record layout");

}  // namespace

class PredictionStore {
public:
    explicit PredictionStore(const std::string& path) : path_(path) {
        fd_ = ::open(path.c_str(), O_RDWR | O_CREAT, 0666);
        if (fd_ < 0) throw std::runtime_error("store: cannot open " + path);
        struct stat st{};
        fstat(fd_, &st);
        const bool fresh = st.st_size == 0;
        if (fresh && ftruncate(fd_, HDR_SIZE + INITIAL_CAP) != 0)
            throw std::runtime_error("store: ftruncate failed");
        remap();
        if (fresh) {
            auto* h = hdr();
            pthread_mutexattr_t at;
            pthread_mutexattr_init(&at);
            pthread_mutexattr_setpshared(&at, PTHREAD_PROCESS_SHARED);
            pthread_mutexattr_setrobust(&at, PTHREAD_MUTEX_ROBUST);
            pthread_mutex_init(&h->mtx, &at);
            pthread_mutexattr_destroy(&at);
            h->committed.store(0);
            h->version = 1;
            std::atomic_thread_fence(std::memory_order_release);
            h->magic = MAGIC;
        } else {
            for (int i = 0; i < 100000 && hdr()->magic != MAGIC; ++i) usleep(10);
            if (hdr()->magic != MAGIC)
                throw std::runtime_error("store: bad magic " + path);
        }
    }
    ~PredictionStore() {
        if (base_) munmap(base_, mapped_);
        if (fd_ >= 0) ::close(fd_);
    }

    void insert(int subject_id, int64_t pred_time_us, float risk_score) {
        auto* h = hdr();
        int rc = pthread_mutex_lock(&h->mtx);
        if (rc == EOWNERDEAD) pthread_mutex_consistent(&h->mtx);
        uint64_t n = h->committed.load(std::memory_order_relaxed);
        while (HDR_SIZE + (n + 1) * sizeof(Rec) > (uint64_t)file_size_) {
            if (ftruncate(fd_, (file_size_ - HDR_SIZE) * 2 + HDR_SIZE) != 0) {
                pthread_mutex_unlock(&h->mtx);
                throw std::runtime_error("store: grow failed");
            }
            remap();
            h = hdr();
        }
        Rec* r = recs() + n;
        r->subject_id = subject_id;
        r->pred_time_us = pred_time_us;
        r->risk_score = risk_score;
        h->committed.store(n + 1, std::memory_order_release);
        pthread_mutex_unlock(&h->mtx);
    }

    // Bulk insert: one lock, contiguous copy (the DP all-gather sink).
    void insert_batch(const std::vector<int>& sids,
                      const std::vector<int64_t>& times,
                      const std::vector<float>& scores) {
        if (sids.size() != times.size() || sids.size() != scores.size())
            throw std::runtime_error("store: batch length mismatch");
        auto* h = hdr();
        int rc = pthread_mutex_lock(&h->mtx);
        if (rc == EOWNERDEAD) pthread_mutex_consistent(&h->mtx);
        uint64_t n = h->committed.load(std::memory_order_relaxed);
        while (HDR_SIZE + (n + sids.size()) * sizeof(Rec) >
               (uint64_t)file_size_) {
            if (ftruncate(fd_, (file_size_ - HDR_SIZE) * 2 + HDR_SIZE) != 0) {
                pthread_mutex_unlock(&h->mtx);
                throw std::runtime_error("store: grow failed");
            }
            remap();
            h = hdr();
        }
        Rec* r = recs() + n;
        for (size_t i = 0; i < sids.size(); ++i) {
            r[i].subject_id = sids[i];
            r[i].pred_time_us = times[i];
            r[i].risk_score = scores[i];
        }
        h->committed.store(n + sids.size(), std::memory_order_release);
        pthread_mutex_unlock(&h->mtx);
    }

    uint64_t count() {
        maybe_remap();
        return hdr()->committed.load(std::memory_order_acquire);
    }

    // Query 2: latest prediction for a patient (backward scan).
    std::optional<std::tuple<int64_t, float>> latest(int subject_id) {
        const uint64_t n = count();
        const Rec* r = recs();
        for (uint64_t i = n; i-- > 0;)
            if (r[i].subject_id == subject_id)
                return std::make_tuple(r[i].pred_time_us, r[i].risk_score);
        return std::nullopt;
    }

    // Query 3: predictions with PRED_TIME >= since ("today's" predictions).
    std::vector<std::tuple<int, int64_t, float>> since(int64_t since_us,
                                                       int limit = 1 << 20) {
        std::vector<std::tuple<int, int64_t, float>> out;
        const uint64_t n = count();
        const Rec* r = recs();
        for (uint64_t i = 0; i < n && (int)out.size() < limit; ++i)
            if (r[i].pred_time_us >= since_us)
                out.emplace_back(r[i].subject_id, r[i].pred_time_us,
                                 r[i].risk_score);
        return out;
    }

    std::vector<std::tuple<int, int64_t, float>> tail(int k) {
        std::vector<std::tuple<int, int64_t, float>> out;
        const uint64_t n = count();
        const Rec* r = recs();
        for (uint64_t i = n > (uint64_t)k ? n - k : 0; i < n; ++i)
            out.emplace_back(r[i].subject_id, r[i].pred_time_us,
                             r[i].risk_score);
        return out;
    }

private:
    StoreHeader* hdr() { return reinterpret_cast<StoreHeader*>(base_); }
    Rec* recs() { return reinterpret_cast<Rec*>((uint8_t*)base_ + HDR_SIZE); }
    void remap() {
        struct stat st{};
        fstat(fd_, &st);
        if (base_) munmap(base_, mapped_);
        file_size_ = st.st_size;
        mapped_ = st.st_size;
        base_ = mmap(nullptr, mapped_, PROT_READ | PROT_WRITE, MAP_SHARED,
                     fd_, 0);
        if (base_ == MAP_FAILED) throw std::runtime_error("store: mmap failed");
    }
    void maybe_remap() {
        struct stat st{};
        fstat(fd_, &st);
        if (st.st_size != (off_t)file_size_) remap();
    }

    std::string path_;
    int fd_ = -1;
    void* base_ = nullptr;
    size_t mapped_ = 0;
    off_t file_size_ = 0;
};

// Query 4: the patients_age table (reference db/init.sql:35-39, ~10k rows).
// Plain in-memory map with file load/save; python side parses the cohort CSV.
class AgeTable {
public:
    AgeTable() = default;

    void set(int subject_id, float age) { ages_[subject_id] = age; }

    void set_many(const std::map<int, float>& m) {
        for (auto& [k, v] : m) ages_[k] = v;
    }

    // Reference fallback: unknown patient -> 65.0 (predictStream.py:151).
    float get(int subject_id, float dflt = 65.0f) const {
        auto it = ages_.find(subject_id);
        return it == ages_.end() ? dflt : it->second;
    }

    bool contains(int subject_id) const { return ages_.count(subject_id) > 0; }
    size_t size() const { return ages_.size(); }

    void save(const std::string& path) const {
        FILE* f = fopen(path.c_str(), "w");
        if (!f) throw std::runtime_error("agetable: cannot write " + path);
        for (auto& [k, v] : ages_) fprintf(f, "%d,%.6f\n", k, v);
        fclose(f);
    }
    void load(const std::string& path) {
        FILE* f = fopen(path.c_str(), "r");
        if (!f) throw std::runtime_error("agetable: cannot read " + path);
        int sid;
        float age;
        while (fscanf(f, "%d,%f\n", &sid, &age) == 2) ages_[sid] = age;
        fclose(f);
    }

private:
    std::map<int, float> ages_;
};

PYBIND11_MODULE(_tskd_store, m) {
    m.doc() = "tskd embedded prediction store + age table (MySQL replacement)";
    py::class_<PredictionStore>(m, "PredictionStore")
        .def(py::init<const std::string&>())
        .def("insert", &PredictionStore::insert, py::arg("subject_id"),
             py::arg("pred_time_us"), py::arg("risk_score"))
        .def("insert_batch", &PredictionStore::insert_batch)
        .def("count", &PredictionStore::count)
        .def("latest", &PredictionStore::latest)
        .def("since", &PredictionStore::since, py::arg("since_us"),
             py::arg("limit") = 1 << 20)
        .def("tail", &PredictionStore::tail);
    py::class_<AgeTable>(m, "AgeTable")
        .def(py::init<>())
        .def("set", &AgeTable::set)
        .def("set_many", &AgeTable::set_many)
        .def("get", &AgeTable::get, py::arg("subject_id"),
             py::arg("dflt") = 65.0f)
        .def("contains", &AgeTable::contains)
        .def("__len__", &AgeTable::size)
        .def("save", &AgeTable::save)
        .def("load", &AgeTable::load);
}
