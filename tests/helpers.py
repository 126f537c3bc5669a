"""Shared test helpers: bridge product part-builder output to the oracle."""
import oracle as o


def oracle_blocks(builder):
    """Build (payload, blocks) for oracle.scan_agg from a product PartBuilder.

    The product descriptor stores header-free streams plus parsed header
    fields (the host parses column headers at load, column.go:331-423); the
    oracle consumes the on-disk column payload WITH its header, so we
    re-materialise the header bytes the parse removed."""
    payload = bytearray(builder.payload)
    blocks = []
    for d in builder.blocks():
        stream = bytes(payload[d.field_off: d.field_off + d.field_len])
        if d.field_enc == 9:    # Plain (null-bearing): the stream already
            col = stream        # carries [type][bytes block], no firstValue
                                # (column.go:266-278 encodeDefault)
        elif d.field_vtype == 3:  # float64: [type][exp BE][first cell] + stream
            col = bytes([d.field_enc]) + int(d.exp).to_bytes(2, "big", signed=True) \
                + o.cell_encode(d.field_first) + stream
        else:                   # int64: [type][first cell] + stream
            col = bytes([d.field_enc]) + o.cell_encode(d.field_first) + stream
        col_off = len(payload)
        payload.extend(col)
        blocks.append(dict(
            series_id=d.series_id, count=d.count,
            ts_enc_with_version=d.ts_enc_with_version, version_enc=d.version_enc,
            ts_min=d.ts_min, ts_max=d.ts_max, version_first=d.version_first,
            ts_off=d.ts_off, ts_len=d.ts_len, ver_len=0,
            col_off=col_off, col_len=len(col),
            tag_off=d.tag_off, tag_len=d.tag_len,
            tag2_off=d.tag2_off, tag2_len=d.tag2_len,
            tag3_off=d.tag3_off, tag3_len=d.tag3_len,
            group_code=d.group_code))
    return bytes(payload), blocks


def oracle_scan(builder, field_vtype, n_groups=1, **kw):
    payload, blocks = oracle_blocks(builder)
    return o.scan_agg(payload, blocks, field_vtype, n_groups=n_groups, **kw)
