// Native CSV loader (the reference's DataVec/CSVRecordReader is a JVM/C++
// component; numpy.loadtxt is ~50x slower than needed for MNIST-scale
// files). Parses comma-delimited float rows into a [rows][cols] fp32 CPU
// tensor with one worker thread per file chunk.

#include <torch/extension.h>

#include <cstdlib>
#include <cstring>
#include <fstream>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

namespace {

void parse_rows(const char* data, const std::vector<size_t>& line_starts,
                size_t row_begin, size_t row_end, int64_t cols, float* out) {
  for (size_t r = row_begin; r < row_end; ++r) {
    const char* p = data + line_starts[r];
    float* dst = out + r * cols;
    for (int64_t c = 0; c < cols; ++c) {
      char* endp = nullptr;
      dst[c] = strtof(p, &endp);
      if (endp == p) {  // empty field / malformed
        dst[c] = 0.f;
        while (*p && *p != ',' && *p != '\n') ++p;
      } else {
        p = endp;
      }
      if (*p == ',') ++p;
    }
  }
}

}  // namespace

torch::Tensor csv_load(const std::string& path, int64_t skip_rows) {
  std::ifstream f(path, std::ios::binary | std::ios::ate);
  TORCH_CHECK(f.good(), "csv_load: cannot open ", path);
  size_t size = (size_t)f.tellg();
  f.seekg(0);
  std::string buf(size, '\0');
  f.read(&buf[0], size);

  // index line starts (skip blank lines)
  std::vector<size_t> line_starts;
  line_starts.reserve(size / 64);
  size_t pos = 0;
  while (pos < size) {
    size_t eol = buf.find('\n', pos);
    if (eol == std::string::npos) eol = size;
    if (eol > pos) line_starts.push_back(pos);
    pos = eol + 1;
  }
  if ((size_t)skip_rows >= line_starts.size())
    return torch::empty({0, 0}, torch::kFloat32);
  line_starts.erase(line_starts.begin(), line_starts.begin() + skip_rows);

  // column count from the first row
  int64_t cols = 1;
  {
    const char* p = buf.data() + line_starts[0];
    while (*p && *p != '\n') {
      if (*p == ',') ++cols;
      ++p;
    }
  }
  int64_t rows = (int64_t)line_starts.size();
  torch::Tensor out = torch::empty({rows, cols}, torch::kFloat32);
  float* optr = out.data_ptr<float>();

  int nthreads = (int)std::min<int64_t>(
      rows, std::max(1u, std::thread::hardware_concurrency()));
  if (nthreads <= 1 || rows < 256) {
    parse_rows(buf.data(), line_starts, 0, rows, cols, optr);
  } else {
    std::vector<std::thread> ts;
    size_t chunk = (rows + nthreads - 1) / nthreads;
    for (int t = 0; t < nthreads; ++t) {
      size_t b = t * chunk, e = std::min<size_t>(b + chunk, rows);
      if (b >= e) break;
      ts.emplace_back(parse_rows, buf.data(), std::cref(line_starts), b, e,
                      cols, optr);
    }
    for (auto& th : ts) th.join();
  }
  return out;
}
