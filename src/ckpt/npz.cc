#include "src/ckpt/npz.h"

#include <zlib.h>

#include <cstdio>
#include <cstring>

namespace nz {

// ------------------------------------------------------------ npy header ---
static std::vector<char> npy_header(const NpyArray& a) {
  std::string descr = a.dtype == DType::F32 ? "<f4" : "<u2";
  std::string dict = "{'descr': '" + descr + "', 'fortran_order': False, 'shape': (";
  for (size_t i = 0; i < a.shape.size(); ++i) {
    dict += std::to_string(a.shape[i]);
    if (a.shape.size() == 1) dict += ",";
    else if (i + 1 < a.shape.size()) dict += ", ";
  }
  dict += "), }";
  // pad with spaces so that 10 + len(dict) + 1 is a multiple of 64
  size_t total = 10 + dict.size() + 1;
  size_t pad = (64 - total % 64) % 64;
  dict += std::string(pad, ' ');
  dict += '\n';
  std::vector<char> h;
  const char magic[] = "\x93NUMPY\x01\x00";
  h.insert(h.end(), magic, magic + 8);
  uint16_t hlen = (uint16_t)dict.size();
  h.push_back(char(hlen & 0xff));
  h.push_back(char(hlen >> 8));
  h.insert(h.end(), dict.begin(), dict.end());
  return h;
}

static bool parse_npy(const std::vector<char>& buf, NpyArray& out) {
  if (buf.size() < 10 || memcmp(buf.data(), "\x93NUMPY", 6) != 0) return false;
  uint8_t major = buf[6];
  size_t hlen, hoff;
  if (major == 1) {
    hlen = (uint8_t)buf[8] | ((uint8_t)buf[9] << 8);
    hoff = 10;
  } else {  // v2: 4-byte header len
    if (buf.size() < 12) return false;
    hlen = (uint8_t)buf[8] | ((uint8_t)buf[9] << 8) | ((uint8_t)buf[10] << 16) |
           ((uint8_t)buf[11] << 24);
    hoff = 12;
  }
  if (hlen > buf.size() || hoff > buf.size() - hlen) return false;  // corrupt
  std::string dict(buf.begin() + hoff, buf.begin() + hoff + hlen);
  auto find_val = [&](const std::string& key) -> std::string {
    auto p = dict.find("'" + key + "'");
    if (p == std::string::npos) return "";
    p = dict.find(':', p);
    return dict.substr(p + 1);
  };
  std::string descr = find_val("descr");
  if (descr.find("<f4") != std::string::npos) out.dtype = DType::F32;
  else if (descr.find("<u2") != std::string::npos) out.dtype = DType::BF16;
  else return false;
  std::string sh = find_val("shape");
  auto l = sh.find('('), r = sh.find(')');
  if (l == std::string::npos || r == std::string::npos) return false;
  out.shape.clear();
  std::string inner = sh.substr(l + 1, r - l - 1);
  size_t pos = 0;
  while (pos < inner.size()) {
    while (pos < inner.size() && !isdigit((unsigned char)inner[pos])) ++pos;
    if (pos >= inner.size()) break;
    long long d = strtoll(inner.c_str() + pos, nullptr, 10);
    if (d < 0 || d > (1ll << 40)) return false;  // corrupt/absurd dim
    out.shape.push_back(d);
    while (pos < inner.size() && isdigit((unsigned char)inner[pos])) ++pos;
  }
  size_t payload = hoff + hlen;
  out.data.assign(buf.begin() + payload, buf.end());
  size_t want = size_t(out.numel()) * dtype_size(out.dtype);
  return out.data.size() == want;
}

// ------------------------------------------------------------- zip write ---
namespace {
struct ZipEntry {
  std::string name;
  uint32_t crc, size, offset;
};
void put16(std::vector<char>& v, uint16_t x) {
  v.push_back(char(x & 0xff)); v.push_back(char(x >> 8));
}
void put32(std::vector<char>& v, uint32_t x) {
  for (int i = 0; i < 4; ++i) v.push_back(char((x >> (8 * i)) & 0xff));
}
}  // namespace

void npz_save(const std::string& path, const std::map<std::string, NpyArray>& arrays) {
  FILE* f = fopen(path.c_str(), "wb");
  NZ_CHECK(f, "cannot open " + path + " for writing");
  std::vector<ZipEntry> entries;
  uint32_t offset = 0;
  for (auto& [name, arr] : arrays) {
    std::string member = name + ".npy";
    std::vector<char> hdr = npy_header(arr);
    uint64_t sz64 = hdr.size() + arr.data.size();
    NZ_CHECK(sz64 < 0xfffffffeull, "npz member too large (zip64 unsupported)");
    uint32_t sz = (uint32_t)sz64;
    uint32_t crc = crc32(0, (const Bytef*)hdr.data(), (uInt)hdr.size());
    crc = crc32(crc, (const Bytef*)arr.data.data(), (uInt)arr.data.size());
    std::vector<char> lh;
    put32(lh, 0x04034b50);
    put16(lh, 20);       // version needed
    put16(lh, 0);        // flags
    put16(lh, 0);        // method: stored
    put16(lh, 0); put16(lh, 0x21);  // dos time/date (fixed -> byte-stable)
    put32(lh, crc);
    put32(lh, sz); put32(lh, sz);
    put16(lh, (uint16_t)member.size());
    put16(lh, 0);        // extra len
    fwrite(lh.data(), 1, lh.size(), f);
    fwrite(member.data(), 1, member.size(), f);
    fwrite(hdr.data(), 1, hdr.size(), f);
    fwrite(arr.data.data(), 1, arr.data.size(), f);
    entries.push_back({member, crc, sz, offset});
    offset += uint32_t(lh.size() + member.size()) + sz;
  }
  uint32_t cd_start = offset, cd_size = 0;
  for (auto& e : entries) {
    std::vector<char> ch;
    put32(ch, 0x02014b50);
    put16(ch, 20); put16(ch, 20);
    put16(ch, 0); put16(ch, 0);
    put16(ch, 0); put16(ch, 0x21);
    put32(ch, e.crc);
    put32(ch, e.size); put32(ch, e.size);
    put16(ch, (uint16_t)e.name.size());
    put16(ch, 0); put16(ch, 0);          // extra, comment
    put16(ch, 0); put16(ch, 0);          // disk, internal attrs
    put32(ch, 0);                        // external attrs
    put32(ch, e.offset);
    fwrite(ch.data(), 1, ch.size(), f);
    fwrite(e.name.data(), 1, e.name.size(), f);
    cd_size += uint32_t(ch.size() + e.name.size());
  }
  std::vector<char> eocd;
  put32(eocd, 0x06054b50);
  put16(eocd, 0); put16(eocd, 0);
  put16(eocd, (uint16_t)entries.size());
  put16(eocd, (uint16_t)entries.size());
  put32(eocd, cd_size);
  put32(eocd, cd_start);
  put16(eocd, 0);
  fwrite(eocd.data(), 1, eocd.size(), f);
  fclose(f);
}

// -------------------------------------------------------------- zip read ---
static uint32_t rd32(const unsigned char* p) {
  return p[0] | (p[1] << 8) | (p[2] << 16) | ((uint32_t)p[3] << 24);
}
static uint16_t rd16(const unsigned char* p) { return p[0] | (p[1] << 8); }

std::map<std::string, NpyArray> npz_load(const std::string& path) {
  FILE* f = fopen(path.c_str(), "rb");
  NZ_CHECK(f, "cannot open " + path);
  fseek(f, 0, SEEK_END);
  long fsize = ftell(f);
  fseek(f, 0, SEEK_SET);
  std::vector<unsigned char> buf(fsize);
  NZ_CHECK(fread(buf.data(), 1, fsize, f) == (size_t)fsize, "short read");
  fclose(f);

  // find EOCD (scan back for signature)
  long eocd = -1;
  for (long i = fsize - 22; i >= 0 && i >= fsize - 22 - 65536; --i) {
    if (rd32(&buf[i]) == 0x06054b50) { eocd = i; break; }
  }
  NZ_CHECK(eocd >= 0, "not a zip file: " + path);
  uint16_t nent = rd16(&buf[eocd + 10]);
  uint32_t cd_off = rd32(&buf[eocd + 16]);

  std::map<std::string, NpyArray> out;
  // every offset below comes from the (untrusted) file: bound-check before
  // dereferencing so a truncated/corrupt checkpoint errors instead of OOB
  auto inb = [&](size_t off, size_t need) {
    NZ_CHECK(off <= (size_t)fsize && need <= (size_t)fsize - off,
             "corrupt zip: offset out of range in " + path);
  };
  size_t p = cd_off;
  for (int e = 0; e < nent; ++e) {
    inb(p, 46);
    NZ_CHECK(rd32(&buf[p]) == 0x02014b50, "bad central directory");
    uint16_t method = rd16(&buf[p + 10]);
    uint32_t csize = rd32(&buf[p + 20]);
    uint32_t crc_expect = rd32(&buf[p + 16]);
    uint16_t nlen = rd16(&buf[p + 28]);
    uint16_t xlen = rd16(&buf[p + 30]);
    uint16_t clen = rd16(&buf[p + 32]);
    uint32_t lho = rd32(&buf[p + 42]);
    inb(p + 46, (size_t)nlen + xlen + clen);
    std::string name((const char*)&buf[p + 46], nlen);
    p += 46 + nlen + xlen + clen;
    NZ_CHECK(method == 0, "npz member " + name + " is compressed (unsupported)");
    // local header
    inb(lho, 30);
    NZ_CHECK(rd32(&buf[lho]) == 0x04034b50, "bad local header");
    uint16_t lnlen = rd16(&buf[lho + 26]);
    uint16_t lxlen = rd16(&buf[lho + 28]);
    size_t data_off = lho + 30 + (size_t)lnlen + lxlen;
    inb(data_off, csize);
    std::vector<char> member(buf.begin() + data_off, buf.begin() + data_off + csize);
    uint32_t crc = crc32(0, (const Bytef*)member.data(), (uInt)member.size());
    NZ_CHECK(crc == crc_expect, "crc mismatch in member " + name);
    NpyArray arr;
    NZ_CHECK(parse_npy(member, arr), "bad npy member " + name);
    std::string key = name;
    if (key.size() > 4 && key.substr(key.size() - 4) == ".npy") key.resize(key.size() - 4);
    out.emplace(key, std::move(arr));
  }
  return out;
}

}  // namespace nz
