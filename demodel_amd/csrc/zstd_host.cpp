// Host reference build of the zstd decoder (shares zstd_common.h with
// csrc/zstd_kernel.hip).  Purpose: debug/verify the format+entropy layer
// on CPU against real encoders without GPU time; the GPU kernel's wave
// orchestration is the only part not exercised here.

#include <pybind11/pybind11.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#include "zstd_common.h"

static int zdbg() {
  static int v = -1;
  if (v < 0) v = getenv("ZSTD_DEBUG") ? 1 : 0;
  return v;
}
#define ZD(...) do { if (zdbg()) fprintf(stderr, __VA_ARGS__); } while (0)

namespace py = pybind11;
using namespace zstd_core;

namespace {

struct HostResult {
  std::string data;
  int64_t status = Z_OK;
  uint64_t consumed = 0;
};

HostResult zstd_decode_host(const uint8_t* src, size_t src_len,
                            size_t cap_hint) {
  HostResult r;
  std::vector<uint8_t> out;
  out.reserve(cap_hint);
  std::vector<uint8_t> lit(140 << 10);

  FReader fr;
  fr.init(src, src_len);
  FseTable* ll_t = new FseTable;
  FseTable* ml_t = new FseTable;
  FseTable* of_t = new FseTable;
  FseTable* scratch = new FseTable;
  HuffState* huf = new HuffState;
  int have_huf = 0;
  int ll_ok = 0, ml_ok = 0, of_ok = 0;
  uint32_t rep[3];

  auto fail = [&](int64_t st) {
    r.status = st;
    r.consumed = fr.pos;
    r.data.assign((const char*)out.data(), out.size());
    delete ll_t; delete ml_t; delete of_t; delete scratch; delete huf;
    return r;
  };

  while (fr.pos < fr.len) {
    uint32_t magic = fr.u32();
    if (fr.fail) return fail(Z_ERR_UNDERRUN);
    if ((magic & 0xFFFFFFF0u) == 0x184D2A50u) {
      uint32_t sz = fr.u32();
      if (!fr.need(sz)) return fail(Z_ERR_UNDERRUN);
      fr.pos += sz;
      continue;
    }
    if (magic != 0xFD2FB528u) return fail(Z_ERR_MAGIC);
    uint8_t fhd = fr.u8();
    if (fhd & 3) return fail(Z_ERR_DICT);
    int checksum = (fhd >> 2) & 1;
    int single_seg = (fhd >> 5) & 1;
    int fcs_flag = (fhd >> 6) & 3;
    if (!single_seg) (void)fr.u8();
    if (fcs_flag == 0) { if (single_seg) (void)fr.u8(); }
    else if (fcs_flag == 1) (void)fr.u16();
    else if (fcs_flag == 2) (void)fr.u32();
    else (void)fr.u64v();
    if (fr.fail) return fail(Z_ERR_UNDERRUN);
    rep[0] = 1; rep[1] = 4; rep[2] = 8;
    have_huf = 0;
    ll_ok = ml_ok = of_ok = 0;

    int last = 0;
    while (!last) {
      uint32_t bh = fr.u24();
      if (fr.fail) return fail(Z_ERR_UNDERRUN);
      last = bh & 1;
      int btype = (bh >> 1) & 3;
      uint32_t bsize = bh >> 3;
      if (btype == 0) {
        if (!fr.need(bsize)) return fail(Z_ERR_UNDERRUN);
        out.insert(out.end(), fr.p + fr.pos, fr.p + fr.pos + bsize);
        fr.pos += bsize;
        continue;
      }
      if (btype == 1) {
        uint8_t v = fr.u8();
        if (fr.fail) return fail(Z_ERR_UNDERRUN);
        out.insert(out.end(), bsize, v);
        continue;
      }
      if (btype == 3) return fail(Z_ERR_FORMAT);

      if (!fr.need(bsize)) return fail(Z_ERR_UNDERRUN);
      const uint8_t* blk = fr.p + fr.pos;
      uint64_t blen = bsize;
      fr.pos += bsize;
      FReader br_;
      br_.init(blk, blen);
      ZD("block: last=%d type=%d size=%u out_pos=%zu\n", last, btype, bsize, out.size());
      uint8_t lh0 = br_.u8();
      int lit_type = lh0 & 3;
      int size_fmt = (lh0 >> 2) & 3;
      uint64_t regen = 0, csize = 0;
      int n_hstreams = 1;
      if (lit_type == 0 || lit_type == 1) {
        if (size_fmt == 0 || size_fmt == 2)
          regen = lh0 >> 3;
        else if (size_fmt == 1)
          regen = (lh0 >> 4) | ((uint64_t)br_.u8() << 4);
        else
          regen = (lh0 >> 4) | ((uint64_t)br_.u8() << 4)
                  | ((uint64_t)br_.u8() << 12);
      } else {
        uint32_t b1, b2, b3, b4;
        if (size_fmt == 0) {
          n_hstreams = 1;
          b1 = br_.u8(); b2 = br_.u8();
          regen = (lh0 >> 4) | ((b1 & 0x3F) << 4);
          csize = (b1 >> 6) | (b2 << 2);
        } else if (size_fmt == 1) {
          n_hstreams = 4;
          b1 = br_.u8(); b2 = br_.u8();
          regen = (lh0 >> 4) | ((b1 & 0x3F) << 4);
          csize = (b1 >> 6) | (b2 << 2);
        } else if (size_fmt == 2) {
          n_hstreams = 4;
          b1 = br_.u8(); b2 = br_.u8(); b3 = br_.u8();
          regen = (lh0 >> 4) | (b1 << 4) | ((uint64_t)(b2 & 3) << 12);
          csize = (b2 >> 2) | (b3 << 6);
        } else {
          n_hstreams = 4;
          b1 = br_.u8(); b2 = br_.u8(); b3 = br_.u8(); b4 = br_.u8();
          regen = (lh0 >> 4) | (b1 << 4) | ((uint64_t)(b2 & 0x3F) << 12);
          csize = (b2 >> 6) | (b3 << 2) | ((uint64_t)b4 << 10);
        }
      }
      ZD("lits: type=%d fmt=%d regen=%llu csize=%llu nh=%d\n", lit_type, size_fmt, (unsigned long long)regen, (unsigned long long)csize, n_hstreams);
      if (br_.fail || regen > (131 << 10)) { ZD("ERR lit hdr\n"); return fail(Z_ERR_FORMAT); }
      if (lit_type == 0) {
        if (!br_.need(regen)) return fail(Z_ERR_UNDERRUN);
        memcpy(lit.data(), br_.p + br_.pos, regen);
        br_.pos += regen;
      } else if (lit_type == 1) {
        if (!br_.need(1)) return fail(Z_ERR_UNDERRUN);
        memset(lit.data(), br_.u8(), regen);
      } else {
        const uint8_t* hsec = br_.p + br_.pos;
        if (!br_.need(csize)) return fail(Z_ERR_UNDERRUN);
        br_.pos += csize;
        uint64_t hoff = 0;
        if (lit_type == 2) {
          uint8_t wbuf[256];
          uint64_t wcons = 0;
          int nw = huf_read_weights(wbuf, hsec, csize, &wcons, scratch);
          ZD("huf weights: nw=%d wcons=%llu\n", nw, (unsigned long long)wcons);
          if (nw < 0 || !huf_build(huf, wbuf, nw))
            { ZD("ERR huf build\n"); return fail(Z_ERR_FORMAT); }
          ZD("huf log=%d\n", huf->log);
          have_huf = 1;
          hoff = wcons;
        } else if (!have_huf) {
          return fail(Z_ERR_FORMAT);
        }
        const uint8_t* hdata = hsec + hoff;
        uint64_t hlen = csize - hoff;
        const uint8_t* hsrc[4];
        uint64_t hsrc_len[4];
        uint8_t* hdst[4];
        uint64_t hdst_len[4];
        int ns;
        if (n_hstreams == 1) {
          ns = 1;
          hsrc[0] = hdata; hsrc_len[0] = hlen;
          hdst[0] = lit.data(); hdst_len[0] = regen;
        } else {
          if (hlen < 6) return fail(Z_ERR_UNDERRUN);
          uint32_t s1 = hdata[0] | (hdata[1] << 8);
          uint32_t s2 = hdata[2] | (hdata[3] << 8);
          uint32_t s3 = hdata[4] | (hdata[5] << 8);
          uint64_t rest = hlen - 6;
          if ((uint64_t)s1 + s2 + s3 > rest) return fail(Z_ERR_FORMAT);
          uint64_t q = (regen + 3) / 4;
          ns = 4;
          const uint8_t* p1 = hdata + 6;
          hsrc[0] = p1; hsrc_len[0] = s1;
          hsrc[1] = p1 + s1; hsrc_len[1] = s2;
          hsrc[2] = p1 + s1 + s2; hsrc_len[2] = s3;
          hsrc[3] = p1 + s1 + s2 + s3; hsrc_len[3] = rest - s1 - s2 - s3;
          for (int k = 0; k < 4; ++k) {
            hdst[k] = lit.data() + (uint64_t)k * q;
            hdst_len[k] = (k < 3) ? q : regen - 3 * q;
          }
        }
        for (int k = 0; k < ns; ++k) {
          BBits bb;
          bb.init(hsrc[k], (int64_t)hsrc_len[k]);
          if (bb.fail && hdst_len[k] > 0) return fail(Z_ERR_FORMAT);
          int log = huf->log;
          for (uint64_t i = 0; i < hdst_len[k]; ++i) {
            uint32_t v = bb.peek(log);
            uint16_t e = huf->lut[v];
            bb.skip(e & 0xF);
            hdst[k][i] = (uint8_t)(e >> 4);
          }
        }
      }

      // sequences
      uint32_t n_seq;
      uint8_t sb0 = br_.u8();
      if (br_.fail) return fail(Z_ERR_UNDERRUN);
      if (sb0 < 128) n_seq = sb0;  // (trace below)
      else if (sb0 < 255) n_seq = ((uint32_t)(sb0 - 128) << 8) + br_.u8();
      else n_seq = (uint32_t)br_.u16() + 0x7F00;

      ZD("nseq=%u br_pos=%llu blen=%llu\n", n_seq, (unsigned long long)br_.pos, (unsigned long long)br_.len);
      uint64_t lit_used = 0;
      if (n_seq > 0) {
        uint8_t modes = br_.u8();
        if (br_.fail || (modes & 3)) return fail(Z_ERR_FORMAT);
        const uint8_t* tp = br_.p + br_.pos;
        uint64_t tleft = br_.len - br_.pos;
        int used;
        ZD("modes=%02x\n", modes);
        used = seq_table_init(ll_t, &ll_ok, (modes >> 6) & 3, tp, tleft,
                              kLLDefault, 36, 6, 35, 9, nullptr);
        if (used < 0) { ZD("ERR ll table\n"); return fail(Z_ERR_FORMAT); }
        ZD("ll: mode=%d used=%d log=%d\n", (modes>>6)&3, used, ll_t->log);
        tp += used; tleft -= used;
        used = seq_table_init(of_t, &of_ok, (modes >> 4) & 3, tp, tleft,
                              kOFDefault, 29, 5, 31, 8, nullptr);
        if (used < 0) { ZD("ERR of table\n"); return fail(Z_ERR_FORMAT); }
        ZD("of: mode=%d used=%d log=%d\n", (modes>>4)&3, used, of_t->log);
        tp += used; tleft -= used;
        used = seq_table_init(ml_t, &ml_ok, (modes >> 2) & 3, tp, tleft,
                              kMLDefault, 53, 6, 52, 9, nullptr);
        if (used < 0) { ZD("ERR ml table\n"); return fail(Z_ERR_FORMAT); }
        ZD("ml: mode=%d used=%d log=%d\n", (modes>>2)&3, used, ml_t->log);
        tp += used; tleft -= used;

        BBits sq;
        sq.init(tp, (int64_t)tleft);
        if (sq.fail) return fail(Z_ERR_FORMAT);
        uint32_t ll_state = sq.get(ll_t->log);
        uint32_t of_state = sq.get(of_t->log);
        uint32_t ml_state = sq.get(ml_t->log);
        for (uint32_t i = 0; i < n_seq; ++i) {
          uint8_t ofc = of_t->e[of_state].sym;
          uint8_t mlc = ml_t->e[ml_state].sym;
          uint8_t llc = ll_t->e[ll_state].sym;
          if (ofc > 31 || mlc > 52 || llc > 35) { ZD("ERR seq codes %d %d %d\n", ofc, mlc, llc); return fail(Z_ERR_FORMAT); }
          uint32_t ofv = (1u << ofc) + sq.get(ofc);
          uint32_t ml = kMLBase[mlc] + sq.get(kMLExtra[mlc]);
          uint32_t ll = kLLBase[llc] + sq.get(kLLExtra[llc]);
          uint32_t offset;
          if (ofv > 3) {
            offset = ofv - 3;
            rep[2] = rep[1]; rep[1] = rep[0]; rep[0] = offset;
          } else {
            uint32_t idx = ofv + (ll == 0 ? 1 : 0);
            if (idx == 1) {
              offset = rep[0];
            } else if (idx == 2) {
              offset = rep[1];
              rep[1] = rep[0]; rep[0] = offset;
            } else if (idx == 3) {
              offset = rep[2];
              rep[2] = rep[1]; rep[1] = rep[0]; rep[0] = offset;
            } else {
              offset = rep[0] - 1;
              if (rep[0] == 0 || offset == 0) return fail(Z_ERR_FORMAT);
              rep[2] = rep[1]; rep[1] = rep[0]; rep[0] = offset;
            }
          }
          if (i + 1 < n_seq) {
            ll_state = ll_t->e[ll_state].base
                       + sq.get(ll_t->e[ll_state].nbits);
            ml_state = ml_t->e[ml_state].base
                       + sq.get(ml_t->e[ml_state].nbits);
            of_state = of_t->e[of_state].base
                       + sq.get(of_t->e[of_state].nbits);
          }
          if (i < 4 || i + 4 >= n_seq) ZD("seq %u: ll=%u ml=%u off=%u\n", i, ll, ml, offset);
          if (lit_used + ll > regen) { ZD("ERR lit overrun seq %u: used=%llu ll=%u regen=%llu\n", i, (unsigned long long)lit_used, ll, (unsigned long long)regen); return fail(Z_ERR_FORMAT); }
          if (offset > out.size() + ll) { ZD("ERR offset %u > pos %zu seq %u\n", offset, out.size()+ll, i); return fail(Z_ERR_FORMAT); }
          out.insert(out.end(), lit.data() + lit_used,
                     lit.data() + lit_used + ll);
          lit_used += ll;
          for (uint32_t k = 0; k < ml; ++k)
            out.push_back(out[out.size() - offset]);
        }
      }
      if (lit_used > regen) return fail(Z_ERR_FORMAT);
      out.insert(out.end(), lit.data() + lit_used, lit.data() + regen);
    }
    if (checksum) fr.pos += 4;
  }
  r.consumed = fr.pos;
  r.data.assign((const char*)out.data(), out.size());
  delete ll_t; delete ml_t; delete of_t; delete scratch; delete huf;
  return r;
}

}  // namespace

void register_zstd_host(py::module_& m) {
  m.def("zstd_decode",
        [](py::bytes data, size_t cap_hint) {
          std::string s = data;
          HostResult r = zstd_decode_host((const uint8_t*)s.data(),
                                          s.size(), cap_hint);
          return py::make_tuple(py::bytes(r.data), r.status, r.consumed);
        },
        py::arg("data"), py::arg("cap_hint") = 1 << 20,
        "host reference zstd decode -> (bytes, status, consumed)");
}
