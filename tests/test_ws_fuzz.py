"""Fuzz the hand-rolled RFC6455 frame codec (service/ws.py): random
byte soup and adversarial length fields must never raise or over-buffer;
encode->read must round-trip."""
import asyncio
import struct

from hypothesis import given, settings, strategies as st

from aurora_amd.service.ws import (MAX_FRAME_BYTES, encode_frame, mask_frame,
                                   read_frame)


def _feed(data: bytes):
    async def run():
        reader = asyncio.StreamReader()
        reader.feed_data(data)
        reader.feed_eof()
        return await read_frame(reader)
    return asyncio.run(run())


@settings(max_examples=200, deadline=None)
@given(st.binary(max_size=40))
def test_random_bytes_never_raise(data):
    op, payload = _feed(data)
    assert isinstance(op, int) and isinstance(payload, bytes)


@settings(max_examples=100, deadline=None)
@given(st.binary(max_size=200), st.sampled_from([0x1, 0x2, 0x9, 0xA]))
def test_roundtrip_server_and_masked_client_frames(payload, opcode):
    op, out = _feed(encode_frame(payload, opcode))
    assert (op, out) == (opcode, payload)
    op2, out2 = _feed(mask_frame(payload, opcode))
    assert (op2, out2) == (opcode, payload)


def test_giant_length_field_is_refused_without_buffering():
    # 64-bit length of 2**62: must answer close immediately, not buffer
    hdr = bytes([0x81, 127]) + struct.pack(">Q", 1 << 62)
    op, payload = _feed(hdr + b"x" * 64)
    assert op == 0x8 and payload == b""


def test_mid_frame_eof_is_close():
    full = encode_frame(b"hello world", 0x1)
    op, _ = _feed(full[: len(full) - 4])  # truncated payload
    assert op == 0x8
