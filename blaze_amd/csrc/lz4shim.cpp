#include "lz4shim.h"

#include <dlfcn.h>

namespace auron {
namespace {

typedef size_t (*LZ4F_compressBound_t)(size_t, const void*);
typedef size_t (*LZ4F_compressFrame_t)(void*, size_t, const void*, size_t,
                                       const void*);
typedef unsigned (*LZ4F_isError_t)(size_t);
typedef size_t (*LZ4F_createDCtx_t)(void**, unsigned);
typedef size_t (*LZ4F_freeDCtx_t)(void*);
typedef size_t (*LZ4F_decompress_t)(void*, void*, size_t*, const void*, size_t*,
                                    const void*);

struct Lz4Api {
  void* handle = nullptr;
  LZ4F_compressBound_t compressBound;
  LZ4F_compressFrame_t compressFrame;
  LZ4F_isError_t isError;
  LZ4F_createDCtx_t createDCtx;
  LZ4F_freeDCtx_t freeDCtx;
  LZ4F_decompress_t decompress;

  bool init() {
    if (handle) return true;
    void* h = dlopen("liblz4.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("liblz4.so", RTLD_NOW | RTLD_GLOBAL);
    if (!h) return false;
    compressBound = (LZ4F_compressBound_t)dlsym(h, "LZ4F_compressBound");
    compressFrame = (LZ4F_compressFrame_t)dlsym(h, "LZ4F_compressFrame");
    isError = (LZ4F_isError_t)dlsym(h, "LZ4F_isError");
    createDCtx = (LZ4F_createDCtx_t)dlsym(h, "LZ4F_createDecompressionContext");
    freeDCtx = (LZ4F_freeDCtx_t)dlsym(h, "LZ4F_freeDecompressionContext");
    decompress = (LZ4F_decompress_t)dlsym(h, "LZ4F_decompress");
    if (!compressBound || !compressFrame || !isError || !createDCtx ||
        !freeDCtx || !decompress)
      return false;
    handle = h;
    return true;
  }
};

Lz4Api g_lz4;

}  // namespace

bool lz4_compress_frame(const uint8_t* src, size_t len, std::vector<uint8_t>* out,
                        std::string* err) {
  if (!g_lz4.init()) {
    *err = "liblz4.so.1 not available";
    return false;
  }
  size_t bound = g_lz4.compressBound(len, nullptr) + 64;
  size_t base = out->size();
  out->resize(base + bound);
  size_t clen = g_lz4.compressFrame(out->data() + base, bound, src, len, nullptr);
  if (g_lz4.isError(clen)) {
    *err = "LZ4F_compressFrame failed";
    return false;
  }
  out->resize(base + clen);
  return true;
}

bool lz4_decompress_frame(const uint8_t* src, size_t len,
                          std::vector<uint8_t>* out, std::string* err) {
  if (!g_lz4.init()) {
    *err = "liblz4.so.1 not available";
    return false;
  }
  void* ctx = nullptr;
  if (g_lz4.isError(g_lz4.createDCtx(&ctx, 100))) {
    *err = "LZ4F_createDecompressionContext failed";
    return false;
  }
  size_t src_pos = 0;
  uint8_t sink[1 << 16];
  while (src_pos < len) {
    size_t dst_size = sizeof(sink);
    size_t src_size = len - src_pos;
    size_t rc = g_lz4.decompress(ctx, sink, &dst_size, src + src_pos, &src_size,
                                 nullptr);
    if (g_lz4.isError(rc)) {
      g_lz4.freeDCtx(ctx);
      *err = "LZ4F_decompress failed";
      return false;
    }
    out->insert(out->end(), sink, sink + dst_size);
    src_pos += src_size;
    if (src_size == 0 && dst_size == 0) break;
  }
  g_lz4.freeDCtx(ctx);
  return true;
}

// ---- zstd frames -----------------------------------------------------------
namespace {
typedef size_t (*ZSTD_compressBound_t)(size_t);
typedef size_t (*ZSTD_compress_t)(void*, size_t, const void*, size_t, int);
typedef unsigned long long (*ZSTD_getFrameContentLength_t)(const void*,
                                                           size_t);
typedef size_t (*ZSTD_decompress_t)(void*, size_t, const void*, size_t);
typedef unsigned (*ZSTD_isError_t)(size_t);

struct ZstdApi {
  void* handle = nullptr;
  ZSTD_compressBound_t bound = nullptr;
  ZSTD_compress_t compress = nullptr;
  ZSTD_getFrameContentLength_t content_len = nullptr;
  ZSTD_decompress_t decompress = nullptr;
  ZSTD_isError_t iserr = nullptr;
};

ZstdApi& zstd_api() {
  static ZstdApi z;
  if (!z.handle) {
    void* h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("libzstd.so", RTLD_NOW | RTLD_GLOBAL);
    if (h) {
      z.handle = h;
      z.bound = (ZSTD_compressBound_t)dlsym(h, "ZSTD_compressBound");
      z.compress = (ZSTD_compress_t)dlsym(h, "ZSTD_compress");
      z.content_len =
          (ZSTD_getFrameContentLength_t)dlsym(h, "ZSTD_getDecompressedSize");
      z.decompress = (ZSTD_decompress_t)dlsym(h, "ZSTD_decompress");
      z.iserr = (ZSTD_isError_t)dlsym(h, "ZSTD_isError");
    }
  }
  return z;
}
}  // namespace

bool zstd_compress_frame(const uint8_t* src, size_t len, int level,
                         std::vector<uint8_t>* out, std::string* err) {
  ZstdApi& z = zstd_api();
  if (!z.compress || !z.bound || !z.iserr) {
    *err = "libzstd.so.1 not available";
    return false;
  }
  size_t base = out->size();
  size_t cap = z.bound(len);
  out->resize(base + cap);
  size_t rc = z.compress(out->data() + base, cap, src, len, level);
  if (z.iserr(rc)) {
    *err = "zstd compress failed";
    return false;
  }
  out->resize(base + rc);
  return true;
}

bool zstd_decompress_frame(const uint8_t* src, size_t len,
                           std::vector<uint8_t>* out, std::string* err) {
  ZstdApi& z = zstd_api();
  if (!z.decompress || !z.content_len || !z.iserr) {
    *err = "libzstd.so.1 not available";
    return false;
  }
  unsigned long long need = z.content_len(src, len);
  if ((need == 0 && len > 8) || need > (1ull << 33)) {
    *err = "zstd frame without content length";
    return false;
  }
  size_t base = out->size();
  out->resize(base + (size_t)need);
  size_t rc = z.decompress(out->data() + base, (size_t)need, src, len);
  if (z.iserr(rc) || rc != (size_t)need) {
    *err = "zstd decompress failed";
    return false;
  }
  return true;
}

}  // namespace auron
