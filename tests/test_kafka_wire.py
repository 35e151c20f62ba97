"""Kafka wire protocol: client <-> in-process broker over real TCP
framing (Metadata/Fetch/Produce v0, CRC'd message sets), and the
KafkaDataset offset/checkpoint semantics on that transport."""
import json

import torch  # noqa: F401  (environment parity with other tests)

from deeprec_amd.data.kafka import KafkaDataset
from deeprec_amd.data.kafka_wire import (KafkaWireClient, MiniKafkaBroker,
                                         decode_message_set,
                                         encode_message_set)


def test_message_set_roundtrip():
    msgs = [(0, b"alpha"), (1, b"beta"), (2, b"x" * 1000)]
    data = encode_message_set(msgs)
    assert decode_message_set(data) == msgs
    # truncated tail tolerated (fetch may cut the last message)
    assert decode_message_set(data[:-5]) == msgs[:2]


def test_client_metadata_fetch_produce():
    broker = MiniKafkaBroker()
    try:
        broker.seed("clicks", 0, [b"m0", b"m1", b"m2"])
        c = KafkaWireClient(broker.host, broker.port)
        meta = broker and c.metadata(["clicks"])
        assert meta["topics"]["clicks"] == [0]
        assert meta["brokers"][0][2] == broker.port
        msgs, hw = c.fetch("clicks", 0, 0)
        assert hw == 3
        assert [v for _, v in msgs] == [b"m0", b"m1", b"m2"]
        # fetch from a mid offset
        msgs, _ = c.fetch("clicks", 0, 2)
        assert msgs == [(2, b"m2")]
        # produce appends and returns the base offset
        base = c.produce("clicks", 0, [b"m3", b"m4"])
        assert base == 3
        msgs, hw = c.fetch("clicks", 0, 3)
        assert hw == 5 and [v for _, v in msgs] == [b"m3", b"m4"]
        c.close()
    finally:
        broker.close()


def test_kafka_dataset_over_wire(tmp_path):
    broker = MiniKafkaBroker()
    try:
        rows = [json.dumps({"id": i}).encode() for i in range(10)]
        broker.seed("train", 0, rows)
        servers = f"{broker.host}:{broker.port}"
        ds = KafkaDataset(["train:0:0"], servers=servers,
                          message_parser=json.loads, batch_size=4)
        got = [m["id"] for batch in ds for m in batch]
        assert got == list(range(10))
        # offsets checkpoint + resume mid-stream on the wire transport
        ds2 = KafkaDataset(["train:0:0"], servers=servers,
                           message_parser=json.loads, batch_size=3)
        it = iter(ds2)
        first = next(it)
        assert [m["id"] for m in first] == [0, 1, 2]
        ck = tmp_path / "kafka_state.json"
        ds2.save(str(ck))
        ds3 = KafkaDataset(["train:0:0"], servers=servers,
                           message_parser=json.loads, batch_size=100)
        ds3.restore(str(ck))
        rest = next(iter(ds3))
        assert [m["id"] for m in rest] == list(range(3, 10))
        # new messages produced AFTER the checkpoint are picked up
        broker.seed("train", 0, [json.dumps({"id": 99}).encode()])
        ds4 = KafkaDataset(["train:0:0"], servers=servers,
                           message_parser=json.loads, batch_size=100)
        ds4.restore(str(ck))
        rest = next(iter(ds4))
        assert rest[-1]["id"] == 99
    finally:
        broker.close()
