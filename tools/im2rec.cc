// im2rec — pack a listing of files into a RecordIO pack (+ .idx).
//
// Reference parity: tools/im2rec.cc (304 LoC, the reference's only real
// CLI tool).  Byte format identical to dmlc-core recordio + the python
// reader (mxnet_amd/io/recordio.py):
//   u32 magic 0xced7230a | u32 len (low 29 bits) | payload | pad to 4B
// and each payload is IRHeader{u32 flag, f32 label, u64 id, u64 id2}
// followed by the (already-encoded) image bytes.  This build packs the
// file bytes verbatim (decode/resize belongs to the data pipeline's
// transform stage on this stack, not the packer).
//
// Usage: im2rec <listfile> <root> <out-prefix>
//   listfile lines: <id>\t<label>\t<relative-path>
// Produces <out-prefix>.rec and <out-prefix>.idx.
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <fstream>
#include <iostream>
#include <sstream>
#include <string>
#include <vector>

namespace {

constexpr uint32_t kMagic = 0xced7230a;

#pragma pack(push, 1)
struct IRHeader {
  uint32_t flag;
  float label;
  uint64_t id;
  uint64_t id2;
};
#pragma pack(pop)

bool ReadFile(const std::string& path, std::vector<char>* out) {
  std::ifstream f(path, std::ios::binary | std::ios::ate);
  if (!f) return false;
  auto size = f.tellg();
  out->resize(static_cast<size_t>(size));
  f.seekg(0);
  f.read(out->data(), size);
  return bool(f);
}

}  // namespace

int main(int argc, char** argv) {
  if (argc < 4) {
    std::cerr << "usage: im2rec <listfile> <root> <out-prefix>\n"
                 "  listfile: <id>\\t<label>\\t<relpath> per line\n";
    return 1;
  }
  const std::string listfile = argv[1], root = argv[2], prefix = argv[3];
  std::ifstream list(listfile);
  if (!list) {
    std::cerr << "cannot open " << listfile << "\n";
    return 1;
  }
  std::ofstream rec(prefix + ".rec", std::ios::binary);
  std::ofstream idx(prefix + ".idx");
  std::string line;
  size_t count = 0;
  static const char kPad[4] = {0, 0, 0, 0};
  while (std::getline(list, line)) {
    if (line.empty()) continue;
    std::istringstream ss(line);
    uint64_t id;
    float label;
    std::string rel;
    ss >> id >> label;
    std::getline(ss, rel);
    // strip leading whitespace/tab
    size_t b = rel.find_first_not_of(" \t");
    if (b == std::string::npos) continue;
    rel = rel.substr(b);
    std::vector<char> bytes;
    if (!ReadFile(root + "/" + rel, &bytes)) {
      std::cerr << "skip unreadable " << rel << "\n";
      continue;
    }
    IRHeader hdr{0, label, id, 0};
    uint32_t len = static_cast<uint32_t>(sizeof(hdr) + bytes.size());
    uint64_t pos = static_cast<uint64_t>(rec.tellp());
    rec.write(reinterpret_cast<const char*>(&kMagic), 4);
    rec.write(reinterpret_cast<const char*>(&len), 4);
    rec.write(reinterpret_cast<const char*>(&hdr), sizeof(hdr));
    rec.write(bytes.data(), bytes.size());
    uint32_t pad = (4 - len % 4) % 4;
    if (pad) rec.write(kPad, pad);
    idx << id << '\t' << pos << '\n';
    ++count;
  }
  std::cout << "packed " << count << " records into " << prefix << ".rec\n";
  return 0;
}
