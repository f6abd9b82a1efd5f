"""Golden vectors restated AS DATA from the reference's own tests.

Sources (reference = tikv/tikv snapshot at /root/reference, not present at
run time — these literals were transcribed from its test expectations):

- MEMCMP_CASES: components/codec/src/byte.rs:827-1049 test_memcmp_encode_all
  (src, expected asc encoding, expected desc encoding)
- ROW_V2_UNSIGNED / ROW_V2 / ROW_V2_BIG:
  components/tidb_query_datatype/src/codec/row/v2/encoder_for_test.rs:543-608
  (test_encode_unsigned / test_encode / test_encode_big expected bytes)
- CRC64_CHECK: the CRC-64/XZ standard check value for b"123456789"
  (crc64fast 0.1.0 implements CRC-64/XZ; the reference's own checksum tests
  pin only self-consistency — SURVEY.md §8c)
"""

MEMCMP_CASES = [
    (b"", bytes([0, 0, 0, 0, 0, 0, 0, 0, 247]),
     bytes([255] * 8 + [8])),
    (b"\x00", bytes([0, 0, 0, 0, 0, 0, 0, 0, 248]),
     bytes([255] * 8 + [7])),
    (bytes([1, 2, 3]), bytes([1, 2, 3, 0, 0, 0, 0, 0, 250]),
     bytes([254, 253, 252, 255, 255, 255, 255, 255, 5])),
    (bytes([1, 2, 3, 0]), bytes([1, 2, 3, 0, 0, 0, 0, 0, 251]),
     bytes([254, 253, 252, 255, 255, 255, 255, 255, 4])),
    (bytes([1, 2, 3, 4, 5, 6, 7]), bytes([1, 2, 3, 4, 5, 6, 7, 0, 254]),
     bytes([254, 253, 252, 251, 250, 249, 248, 255, 1])),
    (bytes(8), bytes([0] * 8 + [255] + [0] * 8 + [247]),
     bytes([255] * 8 + [0] + [255] * 8 + [8])),
    (bytes([1, 2, 3, 4, 5, 6, 7, 8]),
     bytes([1, 2, 3, 4, 5, 6, 7, 8, 255] + [0] * 8 + [247]),
     bytes([254, 253, 252, 251, 250, 249, 248, 247, 0] + [255] * 8 + [8])),
    (bytes([1, 2, 3, 4, 5, 6, 7, 8, 9]),
     bytes([1, 2, 3, 4, 5, 6, 7, 8, 255, 9, 0, 0, 0, 0, 0, 0, 0, 248]),
     bytes([254, 253, 252, 251, 250, 249, 248, 247, 0, 246,
            255, 255, 255, 255, 255, 255, 255, 7])),
]

# row v2: cols (1, u64::MAX unsigned), (2, -1)
ROW_V2_UNSIGNED = bytes([128, 0, 2, 0, 0, 0, 1, 2, 8, 0, 9, 0,
                         255, 255, 255, 255, 255, 255, 255, 255, 255])

# row v2 BIG: cols (1,1000),(12,2),(335,NULL),(3,3),(8,32767)
ROW_V2_BIG = bytes([128, 1, 4, 0, 1, 0, 1, 0, 0, 0, 3, 0, 0, 0, 8, 0, 0, 0,
                    12, 0, 0, 0, 79, 1, 0, 0, 2, 0, 0, 0, 3, 0, 0, 0, 5, 0,
                    0, 0, 6, 0, 0, 0, 232, 3, 3, 255, 127, 2])

CRC64_CHECK = (b"123456789", 0x995DC9BBDF1939FA)
