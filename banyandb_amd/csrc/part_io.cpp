// banyandb_amd/csrc/part_io.cpp — on-disk part reader/writer.
//
// Produces and consumes the reference's exact part directory layout
// (banyand/measure/part.go:37-56):
//   metadata.json   partMetadata JSON (part_metadata.go:37-45)
//   meta.bin        zstd(1) of concatenated 40-B primaryBlockMetadata
//                   records (primary_metadata.go:60-83)
//   primary.bin     zstd(1) chunks of concatenated blockMetadata.marshal
//                   records, flushed at >128 KiB uncompressed
//                   (block_metadata.go:129-147; block_writer.go:252-257;
//                   measure.go maxUncompressedPrimaryBlockSize)
//   timestamps.bin  per block: ts stream ++ versions stream
//                   (block.go:386-404)
//   fv.bin          per block: field column payload incl. header
//                   (column.go:183-263)
//   <family>.tfm    per block: marshaled columnFamilyMetadata
//                   (column_metadata.go:47-106; block.go:187-215)
//   <family>.tf     per block: tag column payloads
// Marshal forms: dataBlock = varuint offset + varuint size
// (block_metadata.go:48-52); timestampsMetadata = dataBlock ++ u64BE min ++
// u64BE max ++ encodeType ++ varuint versionOffset ++ u64BE versionFirst ++
// versionEncodeType (block_metadata.go:284-293); columnMetadata =
// EncodeBytes(name) ++ valueType ++ dataBlock; blockMetadata = sid u64BE ++
// varuint uncompressedSize ++ varuint count ++ timestamps ++
// varuint nFamilies ++ per family EncodeBytes(name)+dataBlock ++ field
// columnFamilyMetadata (block_metadata.go:129-147).  Typed column names
// carry the reference's "#<suffix>" (column.go:32-62).
#include "../../include/bydb_gpu.h"

#include <cstdio>
#include <cstring>
#include <dlfcn.h>
#include <map>
#include <string>
#include <sys/stat.h>
#include <vector>

namespace partio {

// ---- zstd via dlopen (like encode.cpp; format per RFC 8878) ----
typedef size_t (*zstd_compress_fn)(void *, size_t, const void *, size_t, int);
typedef size_t (*zstd_decompress_fn)(void *, size_t, const void *, size_t);
typedef unsigned long long (*zstd_getsize_fn)(const void *, size_t);
typedef unsigned (*zstd_iserr_fn)(size_t);
static zstd_compress_fn z_compress;
static zstd_decompress_fn z_decompress;
static zstd_getsize_fn z_getsize;
static zstd_iserr_fn z_iserr;

static bool zstd_load() {
    static int loaded = -1;
    if (loaded >= 0) return loaded;
    void *h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("libzstd.so", RTLD_NOW | RTLD_GLOBAL);
    if (h) {
        z_compress = (zstd_compress_fn)dlsym(h, "ZSTD_compress");
        z_decompress = (zstd_decompress_fn)dlsym(h, "ZSTD_decompress");
        z_getsize = (zstd_getsize_fn)dlsym(h, "ZSTD_getFrameContentSize");
        z_iserr = (zstd_iserr_fn)dlsym(h, "ZSTD_isError");
    }
    loaded = (z_compress && z_decompress && z_getsize && z_iserr) ? 1 : 0;
    return loaded;
}

static void u64be(std::vector<uint8_t> &b, uint64_t v) {
    for (int i = 7; i >= 0; i--) b.push_back((uint8_t)(v >> (8 * i)));
}
static void varu(std::vector<uint8_t> &b, uint64_t u) {
    while (u > 0x7f) {
        b.push_back((uint8_t)(0x80u | (u & 0xff)));
        u >>= 7;
    }
    b.push_back((uint8_t)u);
}
static void enc_bytes(std::vector<uint8_t> &b, const std::string &s) {
    varu(b, s.size());
    b.insert(b.end(), s.begin(), s.end());
}
static uint64_t rd_u64be(const uint8_t *p) {
    uint64_t v = 0;
    for (int i = 0; i < 8; i++) v = (v << 8) | p[i];
    return v;
}
static bool rd_varu(const uint8_t *p, size_t len, size_t *pos, uint64_t *out) {
    uint64_t u = 0;
    unsigned sh = 0;
    while (*pos < len) {
        uint8_t c = p[(*pos)++];
        u |= (uint64_t)(c & 0x7f) << sh;
        if (c < 0x80) { *out = u; return true; }
        sh += 7;
    }
    return false;
}

static bool write_file(const std::string &path, const std::vector<uint8_t> &data) {
    FILE *f = fopen(path.c_str(), "wb");
    if (!f) return false;
    bool ok = data.empty() || fwrite(data.data(), 1, data.size(), f) == data.size();
    fclose(f);
    return ok;
}
static bool read_file(const std::string &path, std::vector<uint8_t> &out) {
    FILE *f = fopen(path.c_str(), "rb");
    if (!f) return false;
    fseek(f, 0, SEEK_END);
    long n = ftell(f);
    fseek(f, 0, SEEK_SET);
    out.resize((size_t)n);
    bool ok = n == 0 || fread(out.data(), 1, (size_t)n, f) == (size_t)n;
    fclose(f);
    return ok;
}

static std::vector<uint8_t> zstd1(const std::vector<uint8_t> &src) {
    std::vector<uint8_t> out(src.size() + src.size() / 2 + 512);
    size_t n = z_compress(out.data(), out.size(), src.data(), src.size(), 1);
    out.resize(n);
    return out;
}

// tag column per-row decoded value lengths — needed for the reference's
// uncompressedSizeBytes accounting (block.go:292-320).  Parses the plain
// dictionary form (device-parseable subset).
static bool dict_row_len_sum(const uint8_t *p, size_t len, uint64_t nrows,
                             uint64_t *out_sum) {
    size_t pos = 0;
    uint64_t count;
    if (!rd_varu(p, len, &pos, &count)) return false;
    if (count == 0) { *out_sum = 0; return true; }
    // lengths block (plain only)
    if (pos >= len || p[pos] != 0) return false;
    pos++;
    uint64_t ll = p[pos++];
    const uint8_t *lb = p + pos;
    pos += ll;
    uint8_t wt = lb[0];
    uint32_t wbytes = wt == 0 ? 1 : wt == 1 ? 2 : wt == 2 ? 4 : 8;
    std::vector<uint64_t> vlen(count);
    for (uint64_t v = 0; v < count; v++) {
        uint64_t a = 0;
        for (uint32_t b = 0; b < wbytes; b++) a = (a << 8) | lb[1 + v * wbytes + b];
        vlen[v] = a == 0 ? 0 : a - 1;
    }
    // skip values payload (plain only)
    if (pos >= len || p[pos] != 0) return false;
    pos++;
    uint64_t vl = p[pos++];
    pos += vl;
    // bitpack RLE: [32b entries][8b width][..] MSB-first
    const uint8_t *rp = p + pos;
    uint32_t entries = (uint32_t)((rp[0] << 24) | (rp[1] << 16) | (rp[2] << 8) | rp[3]);
    uint32_t width = rp[4];
    uint64_t sum = 0, rows = 0;
    uint64_t bit = 40;
    auto rd_bits = [&](uint32_t n) {
        uint64_t byte = bit >> 3;
        uint32_t sh = (uint32_t)(bit & 7);
        uint64_t acc = 0;
        for (int i = 0; i < 8; i++) acc = (acc << 8) | rp[byte + (uint64_t)i];
        bit += n;
        return (acc >> (64 - sh - n)) & ((1ull << n) - 1);
    };
    for (uint32_t e = 0; e + 1 < entries; e += 2) {
        uint64_t code = rd_bits(width);
        uint64_t cnt = rd_bits(width);
        if (code < count) sum += vlen[code] * cnt;
        rows += cnt;
    }
    if (rows != nrows) return false;
    *out_sum = sum;
    return true;
}

}  // namespace partio

using namespace partio;

// Write the builder's blocks as a reference-layout part directory.
// field_name: the measure field's base name (suffix added per value type);
// tag_family/tag_names: the dictionary tag columns (slot order).
extern "C" int bydb_part_write_dir(bydb_part_builder *b, const char *path,
                                   const char *field_name,
                                   const char *tag_family,
                                   const char *const *tag_names, int n_tags) {
    if (!zstd_load()) return BYDB_ERR_BAD_DATA;
    const uint8_t *payload = bydb_part_builder_payload(b);
    const bydb_block_desc *descs = bydb_part_builder_blocks(b);
    int64_t n_blocks = bydb_part_builder_n_blocks(b);
    if (n_blocks == 0) return BYDB_ERR_BAD_ARG;
    mkdir(path, 0755);
    std::string dir(path);
    std::string fam = tag_family ? tag_family : "default";

    std::vector<uint8_t> ts_bin, fv_bin, tf_bin, tfm_bin;
    std::vector<uint8_t> primary_chunk, primary_bin, meta_records;
    uint64_t total_uncompressed = 0, total_count = 0;
    int64_t min_ts = INT64_MAX, max_ts = INT64_MIN;
    uint64_t chunk_sid_first = 0;
    int64_t chunk_min_ts = 0, chunk_max_ts = 0;
    bool chunk_open = false;

    auto flush_chunk = [&]() {
        if (primary_chunk.empty()) return;
        std::vector<uint8_t> comp = zstd1(primary_chunk);
        // primaryBlockMetadata (primary_metadata.go:60-68)
        u64be(meta_records, chunk_sid_first);
        u64be(meta_records, (uint64_t)chunk_min_ts);
        u64be(meta_records, (uint64_t)chunk_max_ts);
        u64be(meta_records, (uint64_t)primary_bin.size());
        u64be(meta_records, (uint64_t)comp.size());
        primary_bin.insert(primary_bin.end(), comp.begin(), comp.end());
        primary_chunk.clear();
        chunk_open = false;
    };

    for (int64_t i = 0; i < n_blocks; i++) {
        const bydb_block_desc &d = descs[i];
        uint64_t n = d.count;
        total_count += n;
        if (d.ts_min < min_ts) min_ts = d.ts_min;
        if (d.ts_max > max_ts) max_ts = d.ts_max;
        // timestamps.bin — the builder writes ts stream, versions stream
        // and field stream contiguously (add_block_common), so the
        // versions length is the gap up to the field stream
        uint64_t ts_off = ts_bin.size();
        uint64_t ver_len = d.field_off - (d.ts_off + d.ts_len);
        uint64_t ts_size = d.ts_len + ver_len;
        ts_bin.insert(ts_bin.end(), payload + d.ts_off,
                      payload + d.ts_off + ts_size);
        // fv.bin — field column payload with header
        uint64_t fv_off = fv_bin.size();
        const char *suffix = d.field_vtype == BYDB_VT_FLOAT64 ? "#float" : "#int";
        fv_bin.push_back(d.field_enc);
        if (d.field_vtype == BYDB_VT_FLOAT64) {
            fv_bin.push_back((uint8_t)((uint16_t)d.exp >> 8));
            fv_bin.push_back((uint8_t)((uint16_t)d.exp & 0xff));
        }
        {   // firstValue as order-preserving cell (convert/number.go:33-46)
            int64_t v = d.field_first;
            uint64_t u = v >= 0 ? ((uint64_t)v | (1ULL << 63))
                                : ((1ULL << 63) - (uint64_t)(-(uint64_t)v));
            for (int k = 7; k >= 0; k--) fv_bin.push_back((uint8_t)(u >> (8 * k)));
        }
        fv_bin.insert(fv_bin.end(), payload + d.field_off,
                      payload + d.field_off + d.field_len);
        uint64_t fv_size = fv_bin.size() - fv_off;
        // tag columns -> <family>.tf + columnFamilyMetadata -> .tfm
        uint64_t tfm_off = tfm_bin.size(), tfm_size = 0;
        uint64_t tag_cells = 0;
        int written_tags = 0;
        {
            std::vector<uint8_t> cfm;
            uint64_t offs[3], lens[3];
            const uint64_t toffs[3] = {d.tag_off, d.tag2_off, d.tag3_off};
            const uint64_t tlens[3] = {d.tag_len, d.tag2_len, d.tag3_len};
            for (int t = 0; t < n_tags && t < 3; t++) {
                if (tlens[t] == 0) continue;
                offs[written_tags] = tf_bin.size();
                tf_bin.insert(tf_bin.end(), payload + toffs[t],
                              payload + toffs[t] + tlens[t]);
                lens[written_tags] = tlens[t];
                uint64_t cell_sum = 0;
                if (payload[toffs[t]] == BYDB_ENC_DICTIONARY &&
                    dict_row_len_sum(payload + toffs[t] + 1, tlens[t] - 1, n,
                                     &cell_sum))
                    tag_cells += cell_sum;
                written_tags++;
            }
            if (written_tags > 0) {
                varu(cfm, (uint64_t)written_tags);
                int wt = 0;
                for (int t = 0; t < n_tags && t < 3; t++) {
                    if (tlens[t] == 0) continue;
                    enc_bytes(cfm, std::string(tag_names[t]) + "#str");
                    cfm.push_back(1);  // ValueTypeStr
                    varu(cfm, offs[wt]);
                    varu(cfm, lens[wt]);
                    wt++;
                }
                tfm_bin.insert(tfm_bin.end(), cfm.begin(), cfm.end());
                tfm_size = cfm.size();
            }
        }
        // uncompressedSizeBytes (block.go:292-320)
        uint64_t uncompressed = n * 16;  // ts + version
        if (written_tags > 0) {
            uncompressed += fam.size();
            for (int t = 0; t < n_tags && t < 3; t++)
                if ((t == 0 ? d.tag_len : t == 1 ? d.tag2_len : d.tag3_len) != 0)
                    uncompressed += strlen(tag_names[t]) + 4;  // "#str"
            uncompressed += tag_cells;
        }
        std::string fname = std::string(field_name) + suffix;
        uncompressed += fname.size() + n * 8;
        total_uncompressed += uncompressed;
        // blockMetadata.marshal (block_metadata.go:129-147)
        std::vector<uint8_t> &pb = primary_chunk;
        if (!chunk_open) {
            chunk_sid_first = d.series_id;
            chunk_min_ts = d.ts_min;
            chunk_max_ts = d.ts_max;
            chunk_open = true;
        }
        if (d.ts_min < chunk_min_ts) chunk_min_ts = d.ts_min;
        if (d.ts_max > chunk_max_ts) chunk_max_ts = d.ts_max;
        u64be(pb, d.series_id);
        varu(pb, uncompressed);
        varu(pb, n);
        // timestampsMetadata: dataBlock ++ min/max ++ enc ++ verOff ++
        // verFirst ++ verEnc (block_metadata.go:284-293)
        varu(pb, ts_off);
        varu(pb, ts_size);
        u64be(pb, (uint64_t)d.ts_min);
        u64be(pb, (uint64_t)d.ts_max);
        pb.push_back(d.ts_enc_with_version);
        varu(pb, d.ts_len);  // versionOffset
        u64be(pb, (uint64_t)d.version_first);
        pb.push_back(d.version_enc);
        // tag families
        varu(pb, written_tags > 0 ? 1u : 0u);
        if (written_tags > 0) {
            enc_bytes(pb, fam);
            varu(pb, tfm_off);
            varu(pb, tfm_size);
        }
        // field columnFamilyMetadata
        varu(pb, 1);
        enc_bytes(pb, fname);
        pb.push_back(d.field_vtype);
        varu(pb, fv_off);
        varu(pb, fv_size);
        if (primary_chunk.size() > (128 << 10)) flush_chunk();
    }
    flush_chunk();

    std::vector<uint8_t> meta_bin = zstd1(meta_records);
    if (!write_file(dir + "/timestamps.bin", ts_bin) ||
        !write_file(dir + "/fv.bin", fv_bin) ||
        !write_file(dir + "/primary.bin", primary_bin) ||
        !write_file(dir + "/meta.bin", meta_bin))
        return BYDB_ERR;
    if (!tfm_bin.empty()) {
        if (!write_file(dir + "/" + fam + ".tfm", tfm_bin) ||
            !write_file(dir + "/" + fam + ".tf", tf_bin))
            return BYDB_ERR;
    }
    uint64_t compressed = ts_bin.size() + fv_bin.size() + primary_bin.size() +
                          meta_bin.size() + tfm_bin.size() + tf_bin.size();
    char js[512];
    snprintf(js, sizeof js,
             "{\"compressedSizeBytes\":%llu,\"uncompressedSizeBytes\":%llu,"
             "\"totalCount\":%llu,\"blocksCount\":%llu,\"minTimestamp\":%lld,"
             "\"maxTimestamp\":%lld}",
             (unsigned long long)compressed,
             (unsigned long long)total_uncompressed,
             (unsigned long long)total_count, (unsigned long long)n_blocks,
             (long long)min_ts, (long long)max_ts);
    std::vector<uint8_t> jsb(js, js + strlen(js));
    if (!write_file(dir + "/metadata.json", jsb)) return BYDB_ERR;
    return BYDB_OK;
}

extern "C" int bydb_part_builder_append_raw(bydb_part_builder *b,
                                            const uint8_t *data, uint64_t len);
extern "C" int bydb_part_builder_append_desc(bydb_part_builder *b,
                                             const bydb_block_desc *d);

// Read a part directory back into a fresh builder (payload + descs),
// ready for upload.  Walks meta.bin -> primary.bin chunks -> blockMetadata
// records, then copies the per-block stream slices.
extern "C" int bydb_part_read_dir(bydb_part_builder *b, const char *path) {
    if (!zstd_load()) return BYDB_ERR_BAD_DATA;
    std::string dir(path);
    std::vector<uint8_t> meta_bin, primary_bin, ts_bin, fv_bin;
    if (!read_file(dir + "/meta.bin", meta_bin) ||
        !read_file(dir + "/primary.bin", primary_bin) ||
        !read_file(dir + "/timestamps.bin", ts_bin) ||
        !read_file(dir + "/fv.bin", fv_bin))
        return BYDB_ERR;
    // tag families are loaded lazily by the NAME each blockMetadata record
    // carries (block_metadata.go:129-147): a part written with any family
    // name round-trips, and a block referencing a family whose files are
    // missing is a loud error, never a silent tag drop.
    struct FamFiles {
        std::vector<uint8_t> tfm, tf;
    };
    std::map<std::string, FamFiles> fams;
    auto load_family = [&](const std::string &name) -> FamFiles * {
        auto it = fams.find(name);
        if (it != fams.end()) return it->second.tfm.empty() ? nullptr
                                                            : &it->second;
        FamFiles &ff = fams[name];
        if (!read_file(dir + "/" + name + ".tfm", ff.tfm) ||
            !read_file(dir + "/" + name + ".tf", ff.tf) || ff.tfm.empty())
            return nullptr;
        return &ff;
    };
    // decompress meta.bin -> primaryBlockMetadata records
    unsigned long long msz = z_getsize(meta_bin.data(), meta_bin.size());
    std::vector<uint8_t> meta_records(msz);
    if (z_iserr(z_decompress(meta_records.data(), msz, meta_bin.data(),
                             meta_bin.size())))
        return BYDB_ERR_BAD_DATA;
    // for each primary chunk: decompress + walk blockMetadata records
    for (size_t mo = 0; mo + 40 <= meta_records.size(); mo += 40) {
        uint64_t offset = rd_u64be(meta_records.data() + mo + 24);
        uint64_t size = rd_u64be(meta_records.data() + mo + 32);
        if (offset + size > primary_bin.size()) return BYDB_ERR_BAD_DATA;
        unsigned long long csz = z_getsize(primary_bin.data() + offset, size);
        std::vector<uint8_t> chunk(csz);
        if (z_iserr(z_decompress(chunk.data(), csz, primary_bin.data() + offset,
                                 size)))
            return BYDB_ERR_BAD_DATA;
        const uint8_t *p = chunk.data();
        size_t len = chunk.size(), pos = 0;
        while (pos < len) {
            bydb_block_desc d;
            memset(&d, 0, sizeof d);
            d.series_id = rd_u64be(p + pos);
            pos += 8;
            uint64_t uncompressed, count;
            if (!rd_varu(p, len, &pos, &uncompressed)) return BYDB_ERR_BAD_DATA;
            if (!rd_varu(p, len, &pos, &count)) return BYDB_ERR_BAD_DATA;
            d.count = (uint32_t)count;
            uint64_t ts_off, ts_size;
            if (!rd_varu(p, len, &pos, &ts_off)) return BYDB_ERR_BAD_DATA;
            if (!rd_varu(p, len, &pos, &ts_size)) return BYDB_ERR_BAD_DATA;
            d.ts_min = (int64_t)rd_u64be(p + pos);
            pos += 8;
            d.ts_max = (int64_t)rd_u64be(p + pos);
            pos += 8;
            d.ts_enc_with_version = p[pos++];
            uint64_t ver_off;
            if (!rd_varu(p, len, &pos, &ver_off)) return BYDB_ERR_BAD_DATA;
            d.version_first = (int64_t)rd_u64be(p + pos);
            pos += 8;
            d.version_enc = p[pos++];
            // tag families (single family per block supported, matching
            // the writer; more is a loud error, not a partial read)
            uint64_t nfam;
            if (!rd_varu(p, len, &pos, &nfam)) return BYDB_ERR_BAD_DATA;
            if (nfam > 1) return BYDB_ERR_BAD_DATA;
            uint64_t tfm_off = 0, tfm_size = 0;
            std::string blk_fam;
            for (uint64_t f = 0; f < nfam; f++) {
                uint64_t nl;
                if (!rd_varu(p, len, &pos, &nl)) return BYDB_ERR_BAD_DATA;
                if (pos + nl > len) return BYDB_ERR_BAD_DATA;
                blk_fam.assign((const char *)p + pos, nl);
                pos += nl;
                if (!rd_varu(p, len, &pos, &tfm_off)) return BYDB_ERR_BAD_DATA;
                if (!rd_varu(p, len, &pos, &tfm_size)) return BYDB_ERR_BAD_DATA;
            }
            // field columnFamilyMetadata
            uint64_t ncols;
            if (!rd_varu(p, len, &pos, &ncols)) return BYDB_ERR_BAD_DATA;
            uint64_t fv_off = 0, fv_size = 0;
            uint8_t vtype = 0;
            for (uint64_t c = 0; c < ncols; c++) {
                uint64_t nl;
                if (!rd_varu(p, len, &pos, &nl)) return BYDB_ERR_BAD_DATA;
                pos += nl;
                vtype = p[pos++];
                if (!rd_varu(p, len, &pos, &fv_off)) return BYDB_ERR_BAD_DATA;
                if (!rd_varu(p, len, &pos, &fv_size)) return BYDB_ERR_BAD_DATA;
            }
            d.field_vtype = vtype;
            // materialise the block into the builder payload: ts streams,
            // then the field stream with its header parsed off
            // (column.go:331-423 host-side parse)
            uint64_t base = bydb_part_builder_payload_len(b);
            // -- timestamps
            d.ts_off = base;
            d.ts_len = ver_off;
            // append ts payload
            if (ts_off + ts_size > ts_bin.size()) return BYDB_ERR_BAD_DATA;
            bydb_part_builder_append_raw(b, ts_bin.data() + ts_off, ts_size);
            // -- field column: parse header
            if (fv_off + fv_size > fv_bin.size()) return BYDB_ERR_BAD_DATA;
            const uint8_t *fp = fv_bin.data() + fv_off;
            size_t hp = 0;
            d.field_enc = fp[hp++];
            if (vtype == BYDB_VT_FLOAT64) {
                d.exp = (int16_t)(((uint16_t)fp[hp] << 8) | fp[hp + 1]);
                hp += 2;
            }
            {   // firstValue cell decode (convert/number.go:95-108)
                uint64_t u = rd_u64be(fp + hp);
                hp += 8;
                if (fp[hp - 8] >= 128) {
                    d.field_first = (int64_t)(u ^ (1ULL << 63));
                } else {
                    d.field_first = -(int64_t)((1ULL << 63) - u);
                }
            }
            d.field_off = bydb_part_builder_payload_len(b);
            d.field_len = fv_size - hp;
            bydb_part_builder_append_raw(b, fp + hp, d.field_len);
            // -- tag columns from the named family's cfm record
            if (tfm_size > 0) {
                FamFiles *ff = load_family(blk_fam);
                if (!ff || tfm_off + tfm_size > ff->tfm.size())
                    return BYDB_ERR_BAD_DATA;
                const uint8_t *cp = ff->tfm.data() + tfm_off;
                size_t cl = tfm_size, cpos = 0;
                uint64_t ncols2;
                if (!rd_varu(cp, cl, &cpos, &ncols2))
                    return BYDB_ERR_BAD_DATA;
                for (uint64_t c = 0; c < ncols2 && c < 3; c++) {
                    uint64_t nl;
                    if (!rd_varu(cp, cl, &cpos, &nl)) return BYDB_ERR_BAD_DATA;
                    cpos += nl;
                    cpos += 1;  // valueType
                    uint64_t toff, tsize;
                    if (!rd_varu(cp, cl, &cpos, &toff)) return BYDB_ERR_BAD_DATA;
                    if (!rd_varu(cp, cl, &cpos, &tsize)) return BYDB_ERR_BAD_DATA;
                    if (toff + tsize > ff->tf.size()) return BYDB_ERR_BAD_DATA;
                    uint64_t dst_off = bydb_part_builder_payload_len(b);
                    bydb_part_builder_append_raw(
                        b, ff->tf.data() + toff, tsize);
                    if (c == 0) { d.tag_off = dst_off; d.tag_len = tsize; }
                    else if (c == 1) { d.tag2_off = dst_off; d.tag2_len = tsize; }
                    else { d.tag3_off = dst_off; d.tag3_len = tsize; }
                }
            }
            bydb_part_builder_append_desc(b, &d);
        }
    }
    return BYDB_OK;
}
