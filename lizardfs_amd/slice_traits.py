"""Goal::Slice::Type / ChunkPartType algebra — pure-Python mirror of the
reference's EC type system (the plugin surface the engine speaks).

References: goal.h:108-119 (type enum), slice_traits.h:141-209 (ec
namespace), :297-349 (stripe/blocks/length math), chunk_part_type.h:140-240
(16-bit id packing, id = type*64 + part).
"""

K_STANDARD = 0
K_TAPE = 1
K_XOR2 = 2
K_XOR9 = 9
K_EC_FIRST = 10                      # goal.h:118 — $ec(2,1)
K_EC_LAST = K_EC_FIRST + 31 * 32 - 1  # $ec(32,32)

MIN_DATA = 2
MAX_DATA = 32
MIN_PARITY = 1
MAX_PARITY = 32

MAX_PARTS_COUNT = 64                 # chunk_part_type.h:145
BLOCK_SIZE = 65536                   # MFSBLOCKSIZE (MFSCommunication.h)
BLOCKS_IN_CHUNK = 1024               # MFSBLOCKSINCHUNK -> 64 MiB chunks
CHUNK_SIZE = BLOCK_SIZE * BLOCKS_IN_CHUNK


def ec_slice_type(k, m):
    """slice_traits.h:148-151."""
    if not (MIN_DATA <= k <= MAX_DATA and MIN_PARITY <= m <= MAX_PARITY):
        raise ValueError(f"ec({k},{m}) out of range")
    return 32 * (k - MIN_DATA) + (m - MIN_PARITY) + K_EC_FIRST


def is_ec(t):
    return K_EC_FIRST <= t <= K_EC_LAST


def is_xor(t):
    return K_XOR2 <= t <= K_XOR9


def data_parts(t):
    """slice_traits.h:159-161 (EC), :227-235 (general)."""
    if is_ec(t):
        return MIN_DATA + (t - K_EC_FIRST) // 32
    if is_xor(t):
        return t - K_XOR2 + 2
    return 1


def parity_parts(t):
    """slice_traits.h:171-173 (EC), :245-253 (general)."""
    if is_ec(t):
        return MIN_PARITY + (t - K_EC_FIRST) % 32
    if is_xor(t):
        return 1
    return 0


def is_ec2(t):
    """Cauchy-matrix types (slice_traits.h:199-209; mirror of the
    reed_solomon.h:168 condition)."""
    return is_ec(t) and (parity_parts(t) >= 5 or
                         (parity_parts(t) == 4 and data_parts(t) > 20))


def chunk_part_id(slice_type, part):
    """chunk_part_type.h:170-174."""
    assert 0 <= part < MAX_PARTS_COUNT
    return slice_type * MAX_PARTS_COUNT + part


def chunk_part_slice_type(pid):
    return pid // MAX_PARTS_COUNT


def chunk_part_index(pid):
    return pid % MAX_PARTS_COUNT


def is_data_part(slice_type, part):
    """slice_traits.h:191-197."""
    return part < data_parts(slice_type)


def number_of_blocks(slice_type, part, blocks_in_chunk=BLOCKS_IN_CHUNK):
    """slice_traits.h:311-316."""
    k = data_parts(slice_type)
    dpi = part if is_data_part(slice_type, part) else 0
    return (blocks_in_chunk + (k - dpi - 1)) // k


def chunk_part_length(slice_type, part, chunk_length):
    """slice_traits.h:332-349."""
    k = data_parts(slice_type)
    if k == 1:
        return chunk_length
    full_stripe = chunk_length // (k * BLOCK_SIZE)
    base_len = full_stripe * BLOCK_SIZE
    rest = chunk_length - base_len * k
    dpi = part if is_data_part(slice_type, part) else 0
    part_rest = max(rest - dpi * BLOCK_SIZE, 0)
    part_rest = min(part_rest, BLOCK_SIZE)
    return base_len + part_rest
