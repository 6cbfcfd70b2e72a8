/* Example user compression plugin implementing the dlopen ABI the host
 * compressed path loads (reference quant/quant.c:57-65, Intel DL-comp
 * style): int8 block quantization with error feedback.
 *
 * Wire block: [f32 scale][f32 reserved][int8 x BLOCK] = 264 bytes.
 * Build: make quantplugin  ->  build/libquant_plugin.so
 * Use:   mlsl_quant_params_t{ lib_path=".../libquant_plugin.so",
 *          elem_in_block=256, block_size=264, ...default func names }
 */
#include <math.h>
#include <stddef.h>
#include <stdint.h>
#include <string.h>

#define BLOCK 256
#define WIRE (BLOCK + 8)

/* src f32 in (+ diff residual), dst wire out. src_data_type: 2 = f32
 * (the only type this sample supports). */
int dl_comp_compress_buffer(void* src_buffer, void* dst_buffer, size_t count,
                            void* diff, int src_data_type, size_t comp_ratio,
                            int method) {
    (void)comp_ratio;
    (void)method;
    if (src_data_type != 2) return 1;
    const float* src = (const float*)src_buffer;
    float* d = (float*)diff;
    uint8_t* w = (uint8_t*)dst_buffer;
    size_t nblocks = (count + BLOCK - 1) / BLOCK;
    for (size_t b = 0; b < nblocks; ++b) {
        size_t base = b * BLOCK;
        size_t n = count - base < BLOCK ? count - base : BLOCK;
        float* hdr = (float*)(w + b * WIRE);
        int8_t* payload = (int8_t*)(w + b * WIRE + 8);
        float m = 0.f;
        for (size_t i = 0; i < n; ++i) {
            float v = src[base + i] + (d ? d[base + i] : 0.f);
            float a = fabsf(v);
            if (a > m) m = a;
        }
        float scale = m > 0.f ? m / 127.f : 1.f;
        hdr[0] = scale;
        hdr[1] = 0.f;
        for (size_t i = 0; i < n; ++i) {
            float v = src[base + i] + (d ? d[base + i] : 0.f);
            float q = nearbyintf(v / scale);
            if (q > 127.f) q = 127.f;
            if (q < -127.f) q = -127.f;
            payload[i] = (int8_t)q;
            if (d) d[base + i] = v - q * scale;
        }
        for (size_t i = n; i < BLOCK; ++i) payload[i] = 0;
    }
    return 0;
}

int dl_comp_decompress_buffer(void* src_buffer, void* dst_buffer, size_t count) {
    const uint8_t* w = (const uint8_t*)src_buffer;
    float* out = (float*)dst_buffer;
    size_t nblocks = (count + BLOCK - 1) / BLOCK;
    for (size_t b = 0; b < nblocks; ++b) {
        size_t base = b * BLOCK;
        size_t n = count - base < BLOCK ? count - base : BLOCK;
        float scale = ((const float*)(w + b * WIRE))[0];
        const int8_t* payload = (const int8_t*)(w + b * WIRE + 8);
        for (size_t i = 0; i < n; ++i) out[base + i] = payload[i] * scale;
    }
    return 0;
}

/* inout += in over block_count wire blocks (compressed domain). */
int dl_comp_compressed_buffer_reduce_sum(const void* in_buffer,
                                         void* inout_buffer,
                                         size_t block_count) {
    const uint8_t* iw = (const uint8_t*)in_buffer;
    uint8_t* aw = (uint8_t*)inout_buffer;
    for (size_t b = 0; b < block_count; ++b) {
        float* ahdr = (float*)(aw + b * WIRE);
        float as = ahdr[0];
        float is = ((const float*)(iw + b * WIRE))[0];
        int8_t* ap = (int8_t*)(aw + b * WIRE + 8);
        const int8_t* ip = (const int8_t*)(iw + b * WIRE + 8);
        float m = 0.f;
        for (size_t i = 0; i < BLOCK; ++i) {
            float v = fabsf(ap[i] * as + ip[i] * is);
            if (v > m) m = v;
        }
        float ns = m > 0.f ? m / 127.f : 1.f;
        for (size_t i = 0; i < BLOCK; ++i) {
            float v = ap[i] * as + ip[i] * is;
            float q = nearbyintf(v / ns);
            if (q > 127.f) q = 127.f;
            if (q < -127.f) q = -127.f;
            ap[i] = (int8_t)q;
        }
        ahdr[0] = ns;
    }
    return 0;
}
