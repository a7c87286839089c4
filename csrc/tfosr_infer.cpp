// tfosr_infer — standalone batch-inference CLI (capability parity with the
// reference's JVM-only path: Inference.scala drove libtensorflow JNI over
// TFRecords with no Python, reference Inference.scala:27-79). This is the
// MI355X-native equivalent: libtorch (ROCm) runs a TorchScript export over
// TFRecord inputs and writes JSON lines — no Python interpreter involved.
//
//   tfosr_infer --export_dir EXPORT --input DIR_OR_FILE --feature x \
//               [--shape 1,28,28] [--batch 256] [--output out.jsonl]
//
// The input feature is a FloatList or Int64List per Example; values are
// flattened to [batch, shape...] float32.
#include <torch/script.h>

#include <cstring>
#include <dirent.h>
#include <fstream>
#include <iostream>
#include <sstream>
#include <string>
#include <vector>

namespace tfosr {
struct ScanResult {
  std::string buffer;
  std::vector<std::pair<size_t, size_t>> records;
};
ScanResult scan_file(const std::string&, bool);
}

// --- minimal tf.train.Example parse (float/int64 features) -----------------

static uint64_t read_varint(const uint8_t*& p, const uint8_t* end) {
  uint64_t v = 0;
  int shift = 0;
  while (p < end) {
    uint8_t b = *p++;
    v |= (uint64_t)(b & 0x7F) << shift;
    if (!(b & 0x80)) return v;
    shift += 7;
  }
  return v;
}

// extract the named feature's numeric values (float or int64) as floats
static bool extract_feature(const uint8_t* rec, size_t len,
                            const std::string& want, std::vector<float>& out) {
  const uint8_t* p = rec;
  const uint8_t* end = rec + len;
  while (p < end) {
    uint64_t key = read_varint(p, end);
    if ((key & 7) != 2) return false;
    uint64_t flen = read_varint(p, end);
    const uint8_t* fend = p + flen;
    if ((key >> 3) == 1) {  // Example.features
      const uint8_t* q = p;
      while (q < fend) {  // repeated map entries
        uint64_t ekey = read_varint(q, fend);
        uint64_t elen = read_varint(q, fend);
        const uint8_t* eend = q + elen;
        if ((ekey >> 3) == 1) {
          std::string name;
          const uint8_t* fe_ptr = nullptr;
          size_t fe_len = 0;
          const uint8_t* r = q;
          while (r < eend) {
            uint64_t mkey = read_varint(r, eend);
            uint64_t mlen = read_varint(r, eend);
            if ((mkey >> 3) == 1) name.assign((const char*)r, mlen);
            if ((mkey >> 3) == 2) { fe_ptr = r; fe_len = mlen; }
            r += mlen;
          }
          if (name == want && fe_ptr) {
            // Feature message: field 2 FloatList / 3 Int64List
            const uint8_t* s = fe_ptr;
            const uint8_t* se = fe_ptr + fe_len;
            while (s < se) {
              uint64_t fkey = read_varint(s, se);
              uint64_t flen2 = read_varint(s, se);
              const uint8_t* le = s + flen2;
              int field = fkey >> 3;
              // list message: field 1 = packed values
              const uint8_t* t = s;
              while (t < le) {
                uint64_t lkey = read_varint(t, le);
                if ((lkey & 7) == 2) {
                  uint64_t plen = read_varint(t, le);
                  const uint8_t* pe = t + plen;
                  if (field == 2) {
                    for (; t + 4 <= pe; t += 4) {
                      float f;
                      memcpy(&f, t, 4);
                      out.push_back(f);
                    }
                  } else if (field == 3) {
                    while (t < pe) out.push_back((float)(int64_t)read_varint(t, pe));
                  }
                  t = pe;
                } else if ((lkey & 7) == 5 && field == 2) {
                  float f;
                  memcpy(&f, t, 4);
                  out.push_back(f);
                  t += 4;
                } else {
                  out.push_back((float)(int64_t)read_varint(t, le));
                }
              }
              s = le;
            }
            return true;
          }
        }
        q = eend;
      }
    }
    p = fend;
  }
  return false;
}

static std::vector<std::string> list_inputs(const std::string& path) {
  std::vector<std::string> files;
  DIR* d = opendir(path.c_str());
  if (!d) return {path};
  while (dirent* e = readdir(d)) {
    std::string n = e->d_name;
    if (n.rfind("part-", 0) == 0) files.push_back(path + "/" + n);
  }
  closedir(d);
  std::sort(files.begin(), files.end());
  return files;
}

int main(int argc, char** argv) {
  std::string export_dir, input, feature = "x", output;
  std::vector<int64_t> shape;
  int batch = 256;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&]() { return std::string(argv[++i]); };
    if (a == "--export_dir") export_dir = next();
    else if (a == "--input") input = next();
    else if (a == "--feature") feature = next();
    else if (a == "--output") output = next();
    else if (a == "--batch") batch = std::stoi(next());
    else if (a == "--shape") {
      std::stringstream ss(next());
      std::string tok;
      while (std::getline(ss, tok, ',')) shape.push_back(std::stol(tok));
    }
  }
  if (export_dir.empty() || input.empty()) {
    std::cerr << "usage: tfosr_infer --export_dir D --input TFRECORDS "
                 "[--feature x] [--shape 1,28,28] [--batch N] [--output F]\n";
    return 2;
  }

  torch::jit::script::Module model = torch::jit::load(export_dir + "/model.pt");
  torch::Device device(torch::hasCUDA() && torch::getNumGPUs() > 0
                           ? torch::kCUDA : torch::kCPU);
  model.to(device);
  model.eval();

  std::ostream* out = &std::cout;
  std::ofstream fout;
  if (!output.empty()) {
    fout.open(output);
    out = &fout;
  }

  std::vector<std::vector<float>> rows;
  size_t total = 0;
  auto flush = [&]() {
    if (rows.empty()) return;
    int64_t n = rows.size();
    int64_t d = rows[0].size();
    torch::Tensor x = torch::empty({n, d});
    float* ptr = x.data_ptr<float>();
    for (int64_t i = 0; i < n; ++i)
      memcpy(ptr + i * d, rows[i].data(), d * sizeof(float));
    if (!shape.empty()) {
      std::vector<int64_t> full = {n};
      for (auto s : shape) full.push_back(s);
      x = x.reshape(full);
    }
    torch::NoGradGuard ng;
    torch::Tensor y = model.forward({x.to(device)}).toTensor().cpu();
    auto yf = y.to(torch::kFloat).reshape({n, -1});
    auto acc = yf.accessor<float, 2>();
    for (int64_t i = 0; i < n; ++i) {
      *out << "[";
      for (int64_t j = 0; j < yf.size(1); ++j)
        *out << (j ? "," : "") << acc[i][j];
      *out << "]\n";
    }
    total += n;
    rows.clear();
  };

  for (const auto& file : list_inputs(input)) {
    auto scan = tfosr::scan_file(file, false);
    for (auto& [off, len] : scan.records) {
      std::vector<float> vals;
      if (extract_feature((const uint8_t*)scan.buffer.data() + off, len,
                          feature, vals)) {
        rows.push_back(std::move(vals));
        if ((int)rows.size() >= batch) flush();
      }
    }
  }
  flush();
  std::cerr << "tfosr_infer: wrote " << total << " predictions\n";
  return 0;
}
