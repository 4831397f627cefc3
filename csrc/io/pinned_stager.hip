// C6 — native checkpoint stager: device tensors stream to a shard file
// through PINNED host buffers with hipMemcpyAsync on a DEDICATED side
// stream (the compute stream is never blocked), double-buffered so the
// D2H copy of chunk n+1 overlaps the file write of chunk n.
// (SURVEY §2.2 "Pinned-memory checkpoint stager", §3.5.)
//
// Output format: safetensors (8-byte LE header length + JSON header +
// raw little-endian blobs) — byte-compatible with safetensors.safe_open,
// so the Python load path (scaffold/ckpt/sharded.py) reads these shards
// unchanged.

#include <unistd.h>

#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "../common.h"

namespace {

const char* st_dtype(const torch::Tensor& t) {
  switch (t.scalar_type()) {
    case torch::kBFloat16: return "BF16";
    case torch::kFloat32: return "F32";
    case torch::kFloat16: return "F16";
    case torch::kFloat64: return "F64";
    case torch::kInt64: return "I64";
    case torch::kInt32: return "I32";
    case torch::kInt16: return "I16";
    case torch::kUInt8: return "U8";
    case torch::kInt8: return "I8";
    case torch::kBool: return "BOOL";
    default: TORCH_CHECK(false, "pinned_stager: unsupported dtype ",
                         t.scalar_type());
  }
}

struct PinnedBuf {
  void* host = nullptr;
  hipEvent_t ev = nullptr;
  size_t bytes_valid = 0;
};

}  // namespace

// names/tensors parallel lists; metadata serialized into __metadata__.
void save_shard_pinned(std::string path, std::vector<std::string> names,
                       std::vector<torch::Tensor> tensors,
                       std::vector<std::string> meta_keys,
                       std::vector<std::string> meta_vals) {
  TORCH_CHECK(names.size() == tensors.size());
  const size_t N = names.size();

  // ---- header (offsets known up front)
  std::string hdr = "{";
  size_t off = 0;
  std::vector<size_t> nbytes(N);
  for (size_t i = 0; i < N; ++i) {
    auto& t = tensors[i];
    TORCH_CHECK(t.is_contiguous(), "save_shard_pinned: ", names[i],
                " must be contiguous");
    nbytes[i] = (size_t)t.numel() * t.element_size();
    hdr += "\"" + names[i] + "\":{\"dtype\":\"" + st_dtype(t) +
           "\",\"shape\":[";
    for (long d = 0; d < t.dim(); ++d) {
      if (d) hdr += ",";
      hdr += std::to_string(t.size(d));
    }
    hdr += "],\"data_offsets\":[" + std::to_string(off) + "," +
           std::to_string(off + nbytes[i]) + "]},";
    off += nbytes[i];
  }
  hdr += "\"__metadata__\":{";
  for (size_t i = 0; i < meta_keys.size(); ++i) {
    if (i) hdr += ",";
    hdr += "\"" + meta_keys[i] + "\":\"" + meta_vals[i] + "\"";
  }
  hdr += "}}";
  while (hdr.size() % 8) hdr += ' ';  // 8-byte alignment padding

  FILE* fh = fopen(path.c_str(), "wb");
  TORCH_CHECK(fh, "save_shard_pinned: cannot open ", path);
  const unsigned long long hlen = hdr.size();
  fwrite(&hlen, 8, 1, fh);
  fwrite(hdr.data(), 1, hdr.size(), fh);

  const bool any_cuda = [&] {
    for (auto& t : tensors)
      if (t.is_cuda()) return true;
    return false;
  }();

  constexpr size_t BUF = 64u << 20;  // 64 MiB per pinned buffer
  hipStream_t side = nullptr;
  PinnedBuf bufs[2];
  if (any_cuda) {
    HIP_CHECK(hipStreamCreateWithFlags(&side, hipStreamNonBlocking));
    // Order the D2H copies after everything already enqueued on the
    // compute stream (the optimizer step that produced this state may
    // still be in flight) — otherwise shards can capture torn
    // mid-update param/moment bytes.
    hipEvent_t ordered;
    HIP_CHECK(hipEventCreateWithFlags(&ordered, hipEventDisableTiming));
    HIP_CHECK(hipEventRecord(ordered, scaffold_stream()));
    HIP_CHECK(hipStreamWaitEvent(side, ordered, 0));
    HIP_CHECK(hipEventDestroy(ordered));
    for (auto& b : bufs) {
      HIP_CHECK(hipHostMalloc(&b.host, BUF, hipHostMallocDefault));
      HIP_CHECK(hipEventCreateWithFlags(&b.ev, hipEventDisableTiming));
    }
  }

  // double-buffered: copy(slot^1) is in flight while write(slot) drains
  int slot = 0;
  int pending = -1;  // slot whose event must fire before its write
  auto drain = [&](int s) {
    HIP_CHECK(hipEventSynchronize(bufs[s].ev));
    fwrite(bufs[s].host, 1, bufs[s].bytes_valid, fh);
  };

  for (size_t i = 0; i < N; ++i) {
    auto& t = tensors[i];
    if (!t.is_cuda()) {
      if (pending >= 0) { drain(pending); pending = -1; }
      fwrite(t.data_ptr(), 1, nbytes[i], fh);
      continue;
    }
    const char* src = (const char*)t.data_ptr();
    size_t left = nbytes[i];
    while (left) {
      const size_t chunk = left < BUF ? left : BUF;
      HIP_CHECK(hipMemcpyAsync(bufs[slot].host, src, chunk,
                               hipMemcpyDeviceToHost, side));
      HIP_CHECK(hipEventRecord(bufs[slot].ev, side));
      bufs[slot].bytes_valid = chunk;
      if (pending >= 0) drain(pending);  // overlaps the copy above
      pending = slot;
      slot ^= 1;
      src += chunk;
      left -= chunk;
    }
  }
  if (pending >= 0) drain(pending);
  fflush(fh);
  // durability before the rank-0 manifest commit (atomic layout §5.4)
  TORCH_CHECK(fsync(fileno(fh)) == 0, "fsync failed for ", path);
  fclose(fh);

  if (any_cuda) {
    for (auto& b : bufs) {
      (void)hipEventDestroy(b.ev);
      (void)hipHostFree(b.host);
    }
    (void)hipStreamDestroy(side);
  }
}
