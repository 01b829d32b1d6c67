/*
 * compress.h — hand-declared prototypes for the system liblz4/libzstd
 * runtime sonames (no dev headers in this image; ABIs are stable).
 * The reference calls the same entry points: LZ4_compress_default /
 * LZ4_decompress_safe (columnar_compression.c:78,183), ZSTD_compress /
 * ZSTD_decompress (:105,207). Link with -l:liblz4.so.1 -l:libzstd.so.1.
 */
#ifndef CSTRIPE_COMPRESS_H
#define CSTRIPE_COMPRESS_H

#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* liblz4 (block API) */
int LZ4_compress_default(const char *src, char *dst, int srcSize, int dstCapacity);
int LZ4_decompress_safe(const char *src, char *dst, int compressedSize, int dstCapacity);
int LZ4_compressBound(int inputSize);

/* libzstd (simple API) */
size_t ZSTD_compress(void *dst, size_t dstCapacity,
                     const void *src, size_t srcSize, int compressionLevel);
size_t ZSTD_decompress(void *dst, size_t dstCapacity, const void *src, size_t srcSize);
size_t ZSTD_compressBound(size_t srcSize);
unsigned ZSTD_isError(size_t code);

#ifdef __cplusplus
}
#endif
#endif
