#include "gen.h"

#include <cinttypes>
#include <cstdio>
#include <random>
#include <vector>

#include "part_writer.h"
#include "values.h"

namespace vl {

static const char* kDictValues[8] = {
    // app/vlogsgenerator/main.go:288-297
    "debug", "info", "warn", "error", "fatal", "ERROR", "FATAL", "INFO",
};

uint64_t generate_part(const std::string& dir, const GenConfig& cfg) {
  std::mt19937_64 rng(cfg.seed);
  PartWriter w(dir, 1);
  uint64_t msg_bytes = 0;
  // Selective-filter invariants recorded at generation time with plain
  // std::string::find / value equality — independent of every scan path
  // (oracle, emu, HIP), so a full-size bench assertion against these counts
  // fails for any kernel that fabricates bitmaps.  "ip=77." can only occur
  // in the "; ip=a.b.c.d;" fragment (uuid is hex, pad is lowercase), and as
  // a phrase it needs no trailing token boundary ('.' is a non-token char),
  // so find() count == phrase-match count.
  uint64_t sel_msg_ip77 = 0, sel_dict0_error = 0;

  uint64_t rows_per_stream = cfg.rows / cfg.streams;
  uint64_t extra = cfg.rows % cfg.streams;
  int64_t ts = cfg.start_ts;
  char buf[512];

  for (uint64_t s = 0; s < cfg.streams; s++) {
    uint64_t stream_rows = rows_per_stream + (s < extra ? 1 : 0);
    StreamID sid;
    sid.account_id = 0;
    sid.project_id = 0;
    sid.id_hi = s;  // increasing => blocks stay in streamID order
    sid.id_lo = 0x9e3779b97f4a7c15ULL ^ s;

    uint64_t done = 0;
    while (done < stream_rows) {
      uint64_t n = std::min(cfg.rows_per_block, stream_rows - done);
      std::vector<int64_t> timestamps(n);
      std::vector<InputColumn> cols;
      cols.reserve(16 + cfg.num_const_fields + cfg.num_var_fields +
                   cfg.num_dict_fields);  // references below must stay valid
      auto& msg = cols.emplace_back();
      msg.name = "";  // _msg, canonical name (log_rows.go:508-513)
      msg.values.reserve(n);

      auto& host = cols.emplace_back();
      host.name = "host";
      auto& worker = cols.emplace_back();
      worker.name = "worker_id";
      auto& runid = cols.emplace_back();
      runid.name = "run_id";
      std::vector<InputColumn*> constf, varf, dictf;
      for (int j = 0; j < cfg.num_const_fields; j++) {
        auto& c = cols.emplace_back();
        c.name = "const_" + std::to_string(j);
        constf.push_back(&c);
      }
      for (int j = 0; j < cfg.num_var_fields; j++) {
        auto& c = cols.emplace_back();
        c.name = "var_" + std::to_string(j);
        varf.push_back(&c);
      }
      for (int j = 0; j < cfg.num_dict_fields; j++) {
        auto& c = cols.emplace_back();
        c.name = "dict_" + std::to_string(j);
        dictf.push_back(&c);
      }
      InputColumn *u8c = nullptr, *u16c = nullptr, *u32c = nullptr, *u64c = nullptr,
                  *i64c = nullptr, *fc = nullptr, *ipc = nullptr, *tsc = nullptr;
      if (cfg.extra_typed_fields) {
        u8c = &cols.emplace_back(); u8c->name = "u8_0";
        u16c = &cols.emplace_back(); u16c->name = "u16_0";
        u32c = &cols.emplace_back(); u32c->name = "u32_0";
        u64c = &cols.emplace_back(); u64c->name = "u64_0";
        i64c = &cols.emplace_back(); i64c->name = "i64_0";
        fc = &cols.emplace_back(); fc->name = "float_0";
        ipc = &cols.emplace_back(); ipc->name = "ip_0";
        tsc = &cols.emplace_back(); tsc->name = "timestamp_0";
      }

      for (uint64_t r = 0; r < n; r++) {
        timestamps[r] = ts;
        ts += cfg.ts_step;
        uint32_t ip = uint32_t(rng());
        uint64_t uu1 = rng(), uu2 = rng(), u64v = rng();
        // main.go:240-241 row shape
        int len = snprintf(buf, sizeof(buf),
                           "message for the stream %" PRIu64 " and worker 0; "
                           "ip=%u.%u.%u.%u; uuid=%016" PRIx64 "-%016" PRIx64
                           "; u64=%" PRIu64,
                           s, ip >> 24, (ip >> 16) & 255, (ip >> 8) & 255, ip & 255,
                           uu1, uu2, u64v);
        std::string m(buf, size_t(len));
        // pad to msg_len with a trailing token (north-star 256 B _msg)
        if (m.size() + 6 < cfg.msg_len) {
          m += "; pad=";
          while (m.size() < cfg.msg_len) m += char('a' + (rng() % 26));
        }
        if (m.find("ip=77.") != std::string::npos) sel_msg_ip77++;
        msg_bytes += m.size();
        msg.values.push_back(std::move(m));

        host.values.push_back("host_" + std::to_string(s));
        worker.values.push_back("0");
        runid.values.push_back("run-0000");
        for (int j = 0; j < cfg.num_const_fields; j++) {
          constf[j]->values.push_back("some value " + std::to_string(j) + " " +
                                      std::to_string(s));
        }
        for (int j = 0; j < cfg.num_var_fields; j++) {
          varf[j]->values.push_back("some value " + std::to_string(j) + " " +
                                    std::to_string(rng()));
        }
        for (int j = 0; j < cfg.num_dict_fields; j++) {
          const char* dv = kDictValues[rng() % 8];
          if (j == 0 && dv[0] == 'e') sel_dict0_error++;  // exactly "error"
          dictf[j]->values.push_back(dv);
        }
        if (cfg.extra_typed_fields) {
          u8c->values.push_back(std::to_string(uint8_t(rng())));
          u16c->values.push_back(std::to_string(uint16_t(rng())));
          u32c->values.push_back(std::to_string(uint32_t(rng())));
          u64c->values.push_back(std::to_string(rng()));
          i64c->values.push_back(std::to_string(int64_t(rng())));
          {
            double f = double(rng() % 10001) / 1000.0;  // main.go:269 shape
            std::string v;
            format_float64(v, f);
            fc->values.push_back(std::move(v));
          }
          {
            uint32_t a = uint32_t(rng());
            std::string v;
            format_ipv4(v, a);
            ipc->values.push_back(std::move(v));
          }
          {
            // random iso8601 within 2000..2100 so the column encodes as
            // valueTypeTimestampISO8601
            int64_t nsecs = 946684800000000000LL +
                            int64_t(rng() % 3155760000ULL) * 1000000000LL +
                            int64_t(rng() % 1000) * 1000000LL;
            std::string v;
            format_timestamp_iso8601(v, nsecs);
            tsc->values.push_back(std::move(v));
          }
        }
      }
      w.add_block(sid, timestamps, cols);
      done += n;
    }
  }
  w.finish();
  // manifest last: doubles as a generation-complete marker for cached dirs
  {
    std::string path = dir + "/gen_manifest.json";
    FILE* f = fopen(path.c_str(), "w");
    if (f) {
      fprintf(f,
              "{\"rows\": %" PRIu64 ", \"msg_bytes\": %" PRIu64
              ", \"sel_msg_ip77\": %" PRIu64 ", \"sel_dict0_error\": %" PRIu64
              "}\n",
              cfg.rows, msg_bytes, sel_msg_ip77, sel_dict0_error);
      fclose(f);
    }
  }
  return msg_bytes;
}

}  // namespace vl
