// tskd wfdb — first-party C++ WFDB waveform reader (formats 16, 80, 212).
//
// Replaces wfdb-python's numpy-backed readers: parses .hea
// headers (single- and multi-signal, same-file multiplexed .dat), decodes
// the sample stream and applies gain/baseline scaling to physical units,
// with WFDB invalid-sample sentinels mapped to NaN. Feeds the replay
// producer (reference sendStream.py:46 wfdb.rdrecord).
//
// Multi-segment records (the MIMIC waveform layout, e.g. reference record
// p000194-2112-05-23-14-34.hea: `<record>/<nseg>` master line, `~ <n>` gap
// segments, a `<name>_layout 0` variable-layout header) are stitched into
// one (sig_len, n_sig) array on the layout's channel set; gap segments and
// segments whose .dat is absent (partially mirrored directories) read as
// NaN, matching the NaN-for-missing convention of the numerics path.
//
// Header grammar handled (see reference record
// p000194-2112-05-23-14-34n.hea):
//   <record>[/<nseg>] <nsig> <fs>[/counter] [<nsamp> [<base_time> [<base_date>]]]
//   <file> <fmt>[x][:][+] <gain>[(baseline)][/units] <adcres> <adczero>
//          <initval> <checksum> <blocksize> <description...>
//   (multi-segment body): <segment_name|~> <nsamp>

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cmath>
#include <cstdint>
#include <cstring>
#include <fstream>
#include <sstream>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

namespace {

struct SigSpec {
    std::string file;
    int fmt = 16;
    double gain = 200.0;
    double baseline = 0.0;
    bool baseline_set = false;
    std::string units;
    int adcres = 16;
    int adczero = 0;
    int initval = 0;
    std::string desc;
};

struct SegSpec {
    std::string name;  // "~" = gap
    long nsamp = 0;
};

struct Header {
    std::string record;
    int nsig = 0;
    int nseg = 0;  // 0 = single-segment
    double fs = 250.0;
    long nsamp = 0;
    std::string base_time, base_date;
    std::vector<SigSpec> sigs;
    std::vector<SegSpec> segs;
};

std::string dirname_of(const std::string& p) {
    auto pos = p.find_last_of('/');
    return pos == std::string::npos ? std::string(".") : p.substr(0, pos);
}

Header parse_header(const std::string& hea_path) {
    std::ifstream f(hea_path);
    if (!f) throw std::runtime_error("wfdb: cannot open " + hea_path);
    Header h;
    std::string line;
    bool first = true;
    while (std::getline(f, line)) {
        if (line.empty() || line[0] == '#') continue;
        // strip \r
        while (!line.empty() && (line.back() == '\r' || line.back() == ' '))
            line.pop_back();
        if (line.empty()) continue;
        std::istringstream ss(line);
        if (first) {
            first = false;
            std::string fs_tok;
            ss >> h.record >> h.nsig >> fs_tok >> h.nsamp;
            // record may carry /nseg (multi-segment master header)
            auto slash = h.record.find('/');
            if (slash != std::string::npos) {
                h.nseg = atoi(h.record.substr(slash + 1).c_str());
                h.record = h.record.substr(0, slash);
            }
            if (!fs_tok.empty())
                h.fs = atof(fs_tok.substr(0, fs_tok.find('/')).c_str());
            ss >> h.base_time >> h.base_date;
            continue;
        }
        if (h.nseg > 0) {
            // segment list: <name|~> <nsamp>
            if ((int)h.segs.size() >= h.nseg) break;
            SegSpec sg;
            ss >> sg.name >> sg.nsamp;
            h.segs.push_back(sg);
            continue;
        }
        if ((int)h.sigs.size() >= h.nsig) break;
        SigSpec s;
        std::string fmt_tok, gain_tok;
        ss >> s.file >> fmt_tok >> gain_tok >> s.adcres >> s.adczero
           >> s.initval;
        int checksum, blocksize;
        ss >> checksum >> blocksize;
        std::string rest;
        std::getline(ss, rest);
        while (!rest.empty() && rest.front() == ' ') rest.erase(0, 1);
        s.desc = rest;
        s.fmt = atoi(fmt_tok.c_str());  // strips x/:/+ suffixes
        // gain[(baseline)][/units]
        auto sl = gain_tok.find('/');
        if (sl != std::string::npos) {
            s.units = gain_tok.substr(sl + 1);
            gain_tok = gain_tok.substr(0, sl);
        }
        auto par = gain_tok.find('(');
        if (par != std::string::npos) {
            s.baseline = atof(gain_tok.substr(par + 1).c_str());
            s.baseline_set = true;
            gain_tok = gain_tok.substr(0, par);
        }
        s.gain = atof(gain_tok.c_str());
        if (s.gain == 0) s.gain = 200.0;  // WFDB default
        if (!s.baseline_set) s.baseline = s.adczero;
        h.sigs.push_back(s);
    }
    if (h.nseg > 0) {
        if ((int)h.segs.size() != h.nseg)
            throw std::runtime_error("wfdb: segment count mismatch in " +
                                     hea_path);
    } else if ((int)h.sigs.size() != h.nsig) {
        throw std::runtime_error("wfdb: signal count mismatch in " + hea_path);
    }
    return h;
}

std::vector<uint8_t> read_file(const std::string& path) {
    std::ifstream f(path, std::ios::binary);
    if (!f) throw std::runtime_error("wfdb: cannot open " + path);
    return std::vector<uint8_t>((std::istreambuf_iterator<char>(f)),
                                std::istreambuf_iterator<char>());
}

// Decode a multiplexed .dat holding `nsig` interleaved signals in `fmt`.
// Returns adc[samp][sig] as int32 with INT32_MIN for invalid.
std::vector<int32_t> decode_dat(const std::vector<uint8_t>& raw, int fmt,
                                int nsig, long nsamp) {
    std::vector<int32_t> out((size_t)nsamp * nsig, INT32_MIN);
    const size_t total = (size_t)nsamp * nsig;
    switch (fmt) {
        case 16: {
            const size_t n = std::min(total, raw.size() / 2);
            for (size_t i = 0; i < n; ++i) {
                int16_t v;
                memcpy(&v, raw.data() + i * 2, 2);
                out[i] = (v == -32768) ? INT32_MIN : (int32_t)v;
            }
            break;
        }
        case 80: {
            const size_t n = std::min(total, raw.size());
            for (size_t i = 0; i < n; ++i) {
                const int v = (int)raw[i] - 128;  // offset binary
                out[i] = (v == -128) ? INT32_MIN : v;
            }
            break;
        }
        case 212: {
            // 2 samples in 3 bytes: s0 = b0 | (b1 & 0x0f) << 8,
            //                       s1 = b2 | (b1 & 0xf0) << 4 ; 12-bit 2c
            const size_t npairs = raw.size() / 3;
            for (size_t p = 0; p < npairs; ++p) {
                const uint8_t b0 = raw[p * 3], b1 = raw[p * 3 + 1],
                              b2 = raw[p * 3 + 2];
                int s0 = b0 | ((b1 & 0x0f) << 8);
                int s1 = b2 | ((b1 & 0xf0) << 4);
                if (s0 > 2047) s0 -= 4096;
                if (s1 > 2047) s1 -= 4096;
                if (p * 2 < total) out[p * 2] = (s0 == -2048) ? INT32_MIN : s0;
                if (p * 2 + 1 < total)
                    out[p * 2 + 1] = (s1 == -2048) ? INT32_MIN : s1;
            }
            break;
        }
        default:
            throw std::runtime_error("wfdb: unsupported format " +
                                     std::to_string(fmt));
    }
    return out;
}

// Decode every signal of single-segment header `h` (its .dat files live in
// `dir`) into phys rows [t0, t0 + n) where n = min(h.nsamp, nmax).
// col_of[i] = output column for h.sigs[i] (-1 drops the signal). When
// `missing_ok`, an absent .dat leaves its span NaN instead of throwing.
void decode_into(const Header& h, const std::string& dir,
                 const std::vector<int>& col_of,
                 std::vector<std::vector<double>>& phys, long t0, long nmax,
                 bool missing_ok) {
    const long n = std::min(h.nsamp, nmax);
    for (int i = 0; i < h.nsig;) {
        const std::string file = h.sigs[i].file;
        std::vector<int> group;
        for (int j = 0; j < h.nsig; ++j)
            if (h.sigs[j].file == file) group.push_back(j);
        while (i < h.nsig && h.sigs[i].file == file) ++i;
        bool wanted = false;
        for (int g : group) wanted |= col_of[g] >= 0;
        if (!wanted) continue;
        std::vector<uint8_t> raw;
        try {
            raw = read_file(dir + "/" + file);
        } catch (const std::runtime_error&) {
            if (missing_ok) continue;  // span stays NaN
            throw;
        }
        const int fmt = h.sigs[group[0]].fmt;
        auto adc = decode_dat(raw, fmt, (int)group.size(), n);
        for (size_t gi = 0; gi < group.size(); ++gi) {
            const SigSpec& s = h.sigs[group[gi]];
            const int col = col_of[group[gi]];
            if (col < 0) continue;
            for (long t = 0; t < n; ++t) {
                const int32_t a = adc[(size_t)t * group.size() + gi];
                phys[col][t0 + t] =
                    (a == INT32_MIN) ? NAN : (a - s.baseline) / s.gain;
            }
        }
    }
}

}  // namespace

// rdrecord: read header + signals; optionally select channels by name
// (order of `channel_names` defines output column order, like wfdb-python's
// channel_names argument used by the reference).
py::dict rdrecord(const std::string& record_path,
                  const std::vector<std::string>& channel_names) {
    Header h = parse_header(record_path + ".hea");
    const std::string dir = dirname_of(record_path);

    // Canonical signal list: the record's own for single-segment; the
    // layout segment's (or first real segment's, fixed-layout) otherwise.
    std::vector<SigSpec> canon;
    if (h.nseg == 0) {
        canon = h.sigs;
    } else {
        std::string layout_name;
        for (auto& sg : h.segs)
            if (sg.name != "~" && sg.nsamp == 0) { layout_name = sg.name; break; }
        if (layout_name.empty())
            for (auto& sg : h.segs)
                if (sg.name != "~") { layout_name = sg.name; break; }
        if (layout_name.empty())
            throw std::runtime_error("wfdb: multi-segment record " + h.record +
                                     " has no readable segment");
        canon = parse_header(dir + "/" + layout_name + ".hea").sigs;
    }
    const int ncanon = (int)canon.size();

    std::vector<std::vector<double>> phys(
        ncanon, std::vector<double>(h.nsamp, NAN));
    if (h.nseg == 0) {
        std::vector<int> ident(ncanon);
        for (int i = 0; i < ncanon; ++i) ident[i] = i;
        decode_into(h, dir, ident, phys, 0, h.nsamp, /*missing_ok=*/false);
    } else {
        long t0 = 0;
        for (auto& sg : h.segs) {
            if (t0 >= h.nsamp) break;
            if (sg.name == "~" || sg.nsamp == 0) { t0 += sg.nsamp; continue; }
            Header seg;
            try {
                seg = parse_header(dir + "/" + sg.name + ".hea");
            } catch (const std::runtime_error&) {
                t0 += sg.nsamp;  // unmirrored segment header -> NaN span
                continue;
            }
            // map segment signals onto canonical columns by description
            std::vector<int> col_of(seg.nsig, -1);
            for (int i = 0; i < seg.nsig; ++i)
                for (int c = 0; c < ncanon; ++c)
                    if (seg.sigs[i].desc == canon[c].desc) {
                        col_of[i] = c;
                        break;
                    }
            decode_into(seg, dir, col_of, phys, t0,
                        std::min(sg.nsamp, h.nsamp - t0), /*missing_ok=*/true);
            t0 += sg.nsamp;
        }
    }

    // channel selection by description
    std::vector<int> sel;
    std::vector<std::string> names;
    if (channel_names.empty()) {
        for (int i = 0; i < ncanon; ++i) sel.push_back(i);
    } else {
        for (auto& want : channel_names)
            for (int i = 0; i < ncanon; ++i)
                if (canon[i].desc == want) { sel.push_back(i); break; }
    }
    for (int i : sel) names.push_back(canon[i].desc);

    py::array_t<double> p_signal({(py::ssize_t)h.nsamp,
                                  (py::ssize_t)sel.size()});
    auto buf = p_signal.mutable_unchecked<2>();
    for (py::ssize_t t = 0; t < (py::ssize_t)h.nsamp; ++t)
        for (py::ssize_t c = 0; c < (py::ssize_t)sel.size(); ++c)
            buf(t, c) = phys[sel[c]][t];

    py::dict out;
    out["record_name"] = h.record;
    out["fs"] = h.fs;
    out["n_sig"] = (int)sel.size();
    out["sig_name"] = names;
    py::list units, gains;
    for (int i : sel) {
        units.append(canon[i].units);
        gains.append(canon[i].gain);
    }
    out["units"] = units;
    out["gain"] = gains;
    out["p_signal"] = p_signal;
    out["base_time"] = h.base_time;
    out["base_date"] = h.base_date;
    out["sig_len"] = h.nsamp;
    out["n_seg"] = h.nseg;
    return out;
}

PYBIND11_MODULE(_tskd_wfdb, m) {
    m.doc() = "tskd first-party WFDB reader (fmt 16/80/212, multi-segment)";
    m.def("rdrecord", &rdrecord, py::arg("record_path"),
          py::arg("channel_names") = std::vector<std::string>{});
}
