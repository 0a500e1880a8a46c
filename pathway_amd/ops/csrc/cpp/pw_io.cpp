// pathway_amd native IO scanner (host C++, no GPU dependency).
//
// The reference implements its connector data plane natively
// (src/connectors/data_storage/ + data_format/, Rust): file readers and
// format parsers run outside Python.  This library is the MI355X-framework
// analog for the file-based formats: mmap the file once, scan it with a
// single-pass state machine, and hand Python flat offset arrays it can
// turn into columns with zero per-row Python work.
//
//   pw_scan_lines   — newline scan: line [start,end) offsets
//   pw_scan_csv     — RFC-4180-style CSV: per-field [start,end) offsets +
//                     per-field needs-unquoting flag, fixed column count
//                     taken from the header row
//
// All functions return 0 on success; buffers are caller-allocated numpy
// arrays (int64), sized via the corresponding _count call.

#include <cstdint>
#include <cstdio>
#include <cstring>

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

namespace {

struct MappedFile {
  const char* data = nullptr;
  int64_t size = 0;
  int fd = -1;

  bool open_path(const char* path) {
    fd = ::open(path, O_RDONLY);
    if (fd < 0) return false;
    struct stat st;
    if (fstat(fd, &st) != 0) {
      ::close(fd);
      return false;
    }
    size = st.st_size;
    if (size == 0) {
      data = nullptr;
      return true;
    }
    void* p = mmap(nullptr, (size_t)size, PROT_READ, MAP_PRIVATE, fd, 0);
    if (p == MAP_FAILED) {
      ::close(fd);
      return false;
    }
    data = (const char*)p;
    return true;
  }

  ~MappedFile() {
    if (data) munmap((void*)data, (size_t)size);
    if (fd >= 0) ::close(fd);
  }
};

}  // namespace

extern "C" {

// number of lines (newline-terminated; a trailing partial line counts).
// memchr: glibc's SIMD scan — a branchy byte loop measured 9x slower.
int64_t pw_count_lines(const char* path) {
  MappedFile f;
  if (!f.open_path(path)) return -1;
  int64_t n = 0;
  const char* p = f.data;
  const char* end = f.data + f.size;
  while (p < end) {
    const char* nl = (const char*)memchr(p, '\n', end - p);
    if (!nl) {
      ++n;  // trailing partial line
      break;
    }
    ++n;
    p = nl + 1;
  }
  return n;
}

// starts[i], ends[i] = byte range of line i (without the newline; a
// trailing '\r' is stripped).  Returns the number of lines written.
int64_t pw_scan_lines(const char* path, int64_t* starts, int64_t* ends,
                      int64_t cap) {
  MappedFile f;
  if (!f.open_path(path)) return -1;
  int64_t n = 0;
  const char* base = f.data;
  const char* p = f.data;
  const char* end = f.data + f.size;
  while (p < end) {
    const char* nl = (const char*)memchr(p, '\n', end - p);
    if (n >= cap) return -2;
    if (!nl) {
      starts[n] = p - base;
      ends[n] = f.size;
      ++n;
      break;
    }
    const char* e = nl;
    if (e > p && e[-1] == '\r') --e;
    starts[n] = p - base;
    ends[n] = e - base;
    ++n;
    p = nl + 1;
  }
  return n;
}

// CSV scan (delimiter + '"' quoting, "" escapes inside quotes).
// First pass: count data rows and the column count of the first row.
int pw_csv_shape(const char* path, char delimiter, int64_t* out_rows,
                 int64_t* out_cols) {
  MappedFile f;
  if (!f.open_path(path)) return 1;
  int64_t rows = 0, cols = 0, cur_cols = 1;
  bool in_quotes = false, any = false;
  for (int64_t i = 0; i < f.size; ++i) {
    char c = f.data[i];
    any = true;
    if (in_quotes) {
      if (c == '"') {
        if (i + 1 < f.size && f.data[i + 1] == '"') ++i;
        else in_quotes = false;
      }
    } else if (c == '"') {
      in_quotes = true;
    } else if (c == delimiter) {
      ++cur_cols;
    } else if (c == '\n') {
      if (rows == 0) cols = cur_cols;
      ++rows;
      cur_cols = 1;
      any = false;
    }
  }
  if (any) {
    if (rows == 0) cols = cur_cols;
    ++rows;
  }
  *out_rows = rows;
  *out_cols = cols;
  return 0;
}

// Second pass: per-field offsets.  fields are row-major: row r field c at
// index r*ncols + c.  quoted[i] = 1 when the field was quoted (the caller
// strips the surrounding quotes and un-doubles "" while decoding).
// Rows with a different field count than ncols are skipped (counted in
// *skipped).  Returns rows actually written.
int64_t pw_scan_csv(const char* path, char delimiter, int64_t ncols,
                    int64_t* starts, int64_t* ends, uint8_t* quoted,
                    int64_t cap_rows, int64_t* skipped) {
  MappedFile f;
  if (!f.open_path(path)) return -1;
  int64_t row = 0;
  *skipped = 0;
  int64_t fs[512];
  int64_t fe[512];
  uint8_t fq[512];
  if (ncols > 512) return -3;
  int64_t nf = 0;
  int64_t field_start = 0;
  bool in_quotes = false;
  bool was_quoted = false;
  auto flush_field = [&](int64_t end) {
    if (nf < 512) {
      int64_t s = field_start, e = end;
      if (e > s && f.data[e - 1] == '\r') --e;
      fs[nf] = s;
      fe[nf] = e;
      fq[nf] = was_quoted ? 1 : 0;
    }
    ++nf;
    was_quoted = false;
  };
  auto flush_row = [&]() -> bool {
    if (nf == ncols) {
      if (row >= cap_rows) return false;
      for (int64_t c = 0; c < ncols; ++c) {
        starts[row * ncols + c] = fs[c];
        ends[row * ncols + c] = fe[c];
        quoted[row * ncols + c] = fq[c];
      }
      ++row;
    } else {
      ++*skipped;
    }
    nf = 0;
    return true;
  };
  for (int64_t i = 0; i < f.size; ++i) {
    char c = f.data[i];
    if (in_quotes) {
      if (c == '"') {
        if (i + 1 < f.size && f.data[i + 1] == '"') ++i;
        else in_quotes = false;
      }
    } else if (c == '"') {
      in_quotes = true;
      was_quoted = true;
    } else if (c == delimiter) {
      flush_field(i);
      field_start = i + 1;
    } else if (c == '\n') {
      flush_field(i);
      field_start = i + 1;
      if (!flush_row()) return -2;
    }
  }
  if (field_start < f.size || nf > 0) {
    flush_field(f.size);
    if (!flush_row()) return -2;
  }
  return row;
}

}  // extern "C"

extern "C" {

// Single-pass normalizer: writes every field of every well-formed row
// (exactly ncols fields) into out as unquoted bytes separated by '\0'.
// Python then does ONE decode + ONE split — no per-field interpreter
// work anywhere.  Returns rows written; *out_len gets the bytes used.
int64_t pw_csv_normalize(const char* path, char delimiter, int64_t ncols,
                         char* out, int64_t cap, int64_t* out_len,
                         int64_t* skipped) {
  MappedFile f;
  if (!f.open_path(path)) return -1;
  int64_t rows = 0;
  *skipped = 0;
  char* w = out;
  char* row_start_w = out;
  int64_t nf = 0;
  bool in_quotes = false;
  const char* end = f.data + f.size;
  const char* p = f.data;
  auto cap_left = [&](int64_t need) { return (w - out) + need <= cap; };
  while (p < end) {
    char c = *p;
    if (in_quotes) {
      if (c == '"') {
        if (p + 1 < end && p[1] == '"') {
          if (!cap_left(1)) return -2;
          *w++ = '"';
          ++p;
        } else {
          in_quotes = false;
        }
      } else {
        if (!cap_left(1)) return -2;
        *w++ = c;
      }
    } else if (c == '"') {
      in_quotes = true;
    } else if (c == delimiter) {
      if (!cap_left(1)) return -2;
      *w++ = '\0';
      ++nf;
    } else if (c == '\n') {
      if (w > row_start_w && w[-1] == '\r') --w;
      if (!cap_left(1)) return -2;
      *w++ = '\0';
      ++nf;
      if (nf == ncols) {
        ++rows;
        row_start_w = w;
      } else {
        w = row_start_w;  // malformed row: roll back
        ++*skipped;
      }
      nf = 0;
    } else if (c == '\r' && p + 1 < end && p[1] == '\n') {
      // handled at '\n'
      if (!cap_left(1)) return -2;
      *w++ = '\r';
    } else {
      if (!cap_left(1)) return -2;
      *w++ = c;
    }
    ++p;
  }
  if (w > row_start_w || nf > 0) {
    if (w > row_start_w && w[-1] == '\r') --w;
    if (!cap_left(1)) return -2;
    *w++ = '\0';
    ++nf;
    if (nf == ncols) ++rows;
    else {
      w = row_start_w;
      ++*skipped;
    }
  }
  *out_len = w - out;
  return rows;
}

}  // extern "C"

// --------------------------------------------------------------- LZ4 -----
// LZ4 block-format compressor/decompressor (host C++), used by the
// persistence input-snapshot chunks (reference input_snapshot.rs:5 uses
// lz4 block compression with size-prepended frames).  Standard LZ4 block
// format: sequences of [token][literals][offset(2B LE)][matchlen...];
// greedy 4-byte hash-chain matcher.  Output interoperates with any LZ4
// block decoder.

extern "C" {

int64_t pw_lz4_compress_bound(int64_t n) {
  return n + n / 255 + 16;
}

// returns compressed size, or -1 on overflow of the out buffer
int64_t pw_lz4_compress(const uint8_t* src, int64_t n, uint8_t* dst,
                        int64_t cap) {
  const int MINMATCH = 4;
  const int64_t MFLIMIT = 12;  // last 12 bytes are always literals
  if (n == 0) return 0;
  uint32_t htab[1 << 14];
  memset(htab, 0xFF, sizeof(htab));
  auto hash4 = [](const uint8_t* p) -> uint32_t {
    uint32_t v;
    memcpy(&v, p, 4);
    return (v * 2654435761u) >> 18;  // 14-bit
  };
  int64_t ip = 0, anchor = 0, op = 0;
  const int64_t mflimit = n - MFLIMIT;
  while (ip < mflimit) {
    uint32_t h = hash4(src + ip);
    int64_t ref = (htab[h] == 0xFFFFFFFFu) ? -1 : (int64_t)htab[h];
    htab[h] = (uint32_t)ip;
    if (ref >= 0 && ip - ref <= 0xFFFF &&
        memcmp(src + ref, src + ip, MINMATCH) == 0) {
      // extend match
      int64_t mlen = MINMATCH;
      while (ip + mlen < n - 5 && src[ref + mlen] == src[ip + mlen]) ++mlen;
      int64_t litlen = ip - anchor;
      // token + literals
      int64_t need = 1 + litlen / 255 + 1 + litlen + 2 + mlen / 255 + 1;
      if (op + need >= cap) return -1;
      uint8_t* tok = dst + op++;
      if (litlen >= 15) {
        *tok = 0xF0;
        int64_t rest = litlen - 15;
        while (rest >= 255) {
          dst[op++] = 255;
          rest -= 255;
        }
        dst[op++] = (uint8_t)rest;
      } else {
        *tok = (uint8_t)(litlen << 4);
      }
      memcpy(dst + op, src + anchor, litlen);
      op += litlen;
      // offset
      uint16_t off = (uint16_t)(ip - ref);
      dst[op++] = (uint8_t)(off & 0xFF);
      dst[op++] = (uint8_t)(off >> 8);
      // match length (stored - MINMATCH)
      int64_t mstore = mlen - MINMATCH;
      if (mstore >= 15) {
        *tok |= 0x0F;
        int64_t rest = mstore - 15;
        while (rest >= 255) {
          dst[op++] = 255;
          rest -= 255;
        }
        dst[op++] = (uint8_t)rest;
      } else {
        *tok |= (uint8_t)mstore;
      }
      ip += mlen;
      anchor = ip;
    } else {
      ++ip;
    }
  }
  // final literals
  int64_t litlen = n - anchor;
  int64_t need = 1 + litlen / 255 + 1 + litlen;
  if (op + need > cap) return -1;
  uint8_t* tok = dst + op++;
  if (litlen >= 15) {
    *tok = 0xF0;
    int64_t rest = litlen - 15;
    while (rest >= 255) {
      dst[op++] = 255;
      rest -= 255;
    }
    dst[op++] = (uint8_t)rest;
  } else {
    *tok = (uint8_t)(litlen << 4);
  }
  memcpy(dst + op, src + anchor, litlen);
  op += litlen;
  return op;
}

// returns decompressed size, or -1 on malformed input / overflow
int64_t pw_lz4_decompress(const uint8_t* src, int64_t n, uint8_t* dst,
                          int64_t cap) {
  int64_t ip = 0, op = 0;
  while (ip < n) {
    uint8_t token = src[ip++];
    int64_t litlen = token >> 4;
    if (litlen == 15) {
      uint8_t b;
      do {
        if (ip >= n) return -1;
        b = src[ip++];
        litlen += b;
      } while (b == 255);
    }
    if (ip + litlen > n || op + litlen > cap) return -1;
    memcpy(dst + op, src + ip, litlen);
    ip += litlen;
    op += litlen;
    if (ip >= n) break;  // last sequence has no match
    if (ip + 2 > n) return -1;
    uint16_t off = (uint16_t)(src[ip] | (src[ip + 1] << 8));
    ip += 2;
    if (off == 0 || off > op) return -1;
    int64_t mlen = (token & 0x0F);
    if (mlen == 15) {
      uint8_t b;
      do {
        if (ip >= n) return -1;
        b = src[ip++];
        mlen += b;
      } while (b == 255);
    }
    mlen += 4;
    if (op + mlen > cap) return -1;
    // overlapping copy must be byte-wise
    const uint8_t* m = dst + op - off;
    for (int64_t j = 0; j < mlen; ++j) dst[op + j] = m[j];
    op += mlen;
  }
  return op;
}

}  // extern "C"
