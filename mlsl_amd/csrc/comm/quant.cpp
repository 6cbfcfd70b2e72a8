#include "quant.hpp"

#include <dlfcn.h>

#include <algorithm>
#include <cmath>
#include <cstring>
#include <mutex>
#include <string>
#include <unordered_map>

#include "../core/log.hpp"

namespace mlsl {

namespace {

inline float Bf16F(uint16_t h) {
    uint32_t u = static_cast<uint32_t>(h) << 16;
    float f;
    std::memcpy(&f, &u, 4);
    return f;
}

inline uint16_t F32B(float f) {
    uint32_t u;
    std::memcpy(&u, &f, 4);
    uint32_t lsb = (u >> 16) & 1;
    u += 0x7fffu + lsb;
    return static_cast<uint16_t>(u >> 16);
}

template <typename LOAD, typename STORE>
void QuantizeImpl(LOAD load, STORE store, void* wire, size_t count, size_t block,
                  bool use_err) {
    const size_t nblocks = (count + block - 1) / block;
    uint8_t* w = static_cast<uint8_t*>(wire);
    for (size_t b = 0; b < nblocks; ++b) {
        const size_t base = b * block;
        const size_t n = std::min(block, count - base);
        uint8_t* wb = w + b * (block + 8);
        float* hdr = reinterpret_cast<float*>(wb);
        int8_t* payload = reinterpret_cast<int8_t*>(wb + 8);
        float m = 0.f;
        for (size_t i = 0; i < n; ++i) m = std::max(m, std::fabs(load(base + i, use_err)));
        const float scale = m > 0.f ? m / 127.f : 1.f;
        hdr[0] = scale;
        hdr[1] = 0.f;
        const float inv = 1.f / scale;
        for (size_t i = 0; i < n; ++i) {
            float v = load(base + i, use_err);
            float q = std::nearbyint(v * inv);
            q = std::min(127.f, std::max(-127.f, q));
            payload[i] = static_cast<int8_t>(q);
            if (use_err) store(base + i, v - q * scale);
        }
        for (size_t i = n; i < block; ++i) payload[i] = 0;
    }
}

}  // namespace

void HostQuantize(const void* in, void* err, void* wire, size_t count, size_t block,
                  DataType dt, bool use_err) {
    if (dt == DataType::F32) {
        const float* p = static_cast<const float*>(in);
        float* e = static_cast<float*>(err);
        QuantizeImpl(
            [&](size_t i, bool ue) { return ue ? p[i] + e[i] : p[i]; },
            [&](size_t i, float v) { e[i] = v; }, wire, count, block, use_err);
    } else if (dt == DataType::BF16) {
        const uint16_t* p = static_cast<const uint16_t*>(in);
        uint16_t* e = static_cast<uint16_t*>(err);
        QuantizeImpl(
            [&](size_t i, bool ue) { return ue ? Bf16F(p[i]) + Bf16F(e[i]) : Bf16F(p[i]); },
            [&](size_t i, float v) { e[i] = F32B(v); }, wire, count, block, use_err);
    } else {
        MLSL_THROW("quantization supports f32/bf16 only");
    }
}

void HostDequantize(const void* wire, void* out, size_t count, size_t block, DataType dt) {
    const size_t nblocks = (count + block - 1) / block;
    const uint8_t* w = static_cast<const uint8_t*>(wire);
    for (size_t b = 0; b < nblocks; ++b) {
        const size_t base = b * block;
        const size_t n = std::min(block, count - base);
        const uint8_t* wb = w + b * (block + 8);
        const float scale = reinterpret_cast<const float*>(wb)[0];
        const int8_t* payload = reinterpret_cast<const int8_t*>(wb + 8);
        if (dt == DataType::F32) {
            float* o = static_cast<float*>(out);
            for (size_t i = 0; i < n; ++i) o[base + i] = payload[i] * scale;
        } else if (dt == DataType::BF16) {
            uint16_t* o = static_cast<uint16_t*>(out);
            for (size_t i = 0; i < n; ++i) o[base + i] = F32B(payload[i] * scale);
        } else {
            MLSL_THROW("dequantization supports f32/bf16 only");
        }
    }
}

void HostQuantAccum(void* acc_wire, const void* wire, size_t count, size_t block) {
    const size_t nblocks = (count + block - 1) / block;
    uint8_t* aw = static_cast<uint8_t*>(acc_wire);
    const uint8_t* iw = static_cast<const uint8_t*>(wire);
    for (size_t b = 0; b < nblocks; ++b) {
        const size_t base = b * block;
        const size_t n = std::min(block, count - base);
        uint8_t* ab = aw + b * (block + 8);
        const uint8_t* ib = iw + b * (block + 8);
        float* ahdr = reinterpret_cast<float*>(ab);
        const float as = ahdr[0];
        const float is = reinterpret_cast<const float*>(ib)[0];
        int8_t* ap = reinterpret_cast<int8_t*>(ab + 8);
        const int8_t* ip = reinterpret_cast<const int8_t*>(ib + 8);
        float m = 0.f;
        for (size_t i = 0; i < n; ++i)
            m = std::max(m, std::fabs(ap[i] * as + ip[i] * is));
        const float ns = m > 0.f ? m / 127.f : 1.f;
        const float inv = 1.f / ns;
        for (size_t i = 0; i < n; ++i) {
            float v = ap[i] * as + ip[i] * is;
            float q = std::nearbyint(v * inv);
            ap[i] = static_cast<int8_t>(std::min(127.f, std::max(-127.f, q)));
        }
        ahdr[0] = ns;
    }
}

const QuantPluginApi* LoadQuantPlugin(const QuantParams& qp) {
    if (qp.lib_path.empty()) return nullptr;
    static std::mutex mu;
    static std::unordered_map<std::string, QuantPluginApi> cache;
    std::lock_guard<std::mutex> lk(mu);
    const std::string key = qp.lib_path + "|" + qp.quant_fn;
    auto it = cache.find(key);
    if (it != cache.end()) return &it->second;

    void* h = dlopen(qp.lib_path.c_str(), RTLD_NOW);
    if (!h) MLSL_THROW(std::string("quant plugin dlopen failed: ") + dlerror());
    QuantPluginApi api{};
    api.quant = reinterpret_cast<decltype(api.quant)>(dlsym(h, qp.quant_fn.c_str()));
    api.dequant =
        reinterpret_cast<decltype(api.dequant)>(dlsym(h, qp.dequant_fn.c_str()));
    api.reduce_sum = reinterpret_cast<decltype(api.reduce_sum)>(
        dlsym(h, qp.reduce_fn.c_str()));
    if (!api.quant || !api.dequant || !api.reduce_sum)
        MLSL_THROW("quant plugin missing symbol(s): " + qp.quant_fn + "/" +
                   qp.dequant_fn + "/" + qp.reduce_fn);
    MLSL_LOG(INFO, "quant plugin loaded: %s (host path; device path keeps "
             "built-in CDNA4 kernels)", qp.lib_path.c_str());
    return &cache.emplace(key, api).first->second;
}

}  // namespace mlsl
