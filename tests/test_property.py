"""Property-based tests (hypothesis): protobuf wire codec and serde
round-trips hold for arbitrary inputs — the fuzzing tier of the test pyramid.
"""

import json

from hypothesis import given, settings
from hypothesis import strategies as st

from k8s_dra_driver_gpu_amd.api.serde import from_dict, to_dict
from k8s_dra_driver_gpu_amd.dra import api as dra
from k8s_dra_driver_gpu_amd.dra.protowire import decode_varint, encode_varint
from k8s_dra_driver_gpu_amd.k8s.celselect import Quantity
from k8s_dra_driver_gpu_amd.plugin.checkpoint import (
    CheckpointData,
    ClaimRef,
    PreparedClaim,
    PreparedDevice,
)

text = st.text(max_size=60)
# protobuf strings are UTF-8; surrogates don't round-trip
clean_text = st.text(
    alphabet=st.characters(blacklist_categories=("Cs",)), max_size=60
)


class TestProtowireProperties:
    @given(st.integers(min_value=0, max_value=2**64 - 1))
    def test_varint_round_trip(self, v):
        out, pos = decode_varint(encode_varint(v), 0)
        assert out == v

    @settings(max_examples=200)
    @given(ns=clean_text, name=clean_text, uid=clean_text)
    def test_claim_round_trip(self, ns, name, uid):
        c = dra.Claim(namespace=ns, name=name, uid=uid)
        back = dra.Claim.from_bytes(c.to_bytes())
        assert back.namespace == ns and back.name == name and back.uid == uid

    @settings(max_examples=100)
    @given(
        devices=st.lists(
            st.tuples(clean_text, st.lists(clean_text, max_size=4)), max_size=5
        ),
        err=clean_text,
    )
    def test_prepare_response_round_trip(self, devices, err):
        resp = dra.NodePrepareResourceResponse(
            devices=[
                dra.Device(device_name=d, cdi_device_ids=ids) for d, ids in devices
            ],
            error=err,
        )
        back = dra.NodePrepareResourceResponse.from_bytes(resp.to_bytes())
        assert back.error == err
        assert [d.device_name for d in back.devices] == [d for d, _ in devices]
        assert [d.cdi_device_ids for d in back.devices] == [ids for _, ids in devices]

    @settings(max_examples=100)
    @given(entries=st.dictionaries(clean_text, clean_text, max_size=6))
    def test_map_round_trip(self, entries):
        resp = dra.NodeUnprepareResourcesResponse()
        for k, v in entries.items():
            resp.claims[k] = dra.NodeUnprepareResourceResponse(error=v)
        back = dra.NodeUnprepareResourcesResponse.from_bytes(resp.to_bytes())
        assert {k: r.error for k, r in back.claims.items()} == entries

    @given(st.binary(max_size=200))
    def test_decoder_never_crashes_unstructured(self, blob):
        """Arbitrary bytes either decode or raise ValueError — never
        anything else (the parser is exposed to the kubelet socket)."""
        try:
            dra.NodePrepareResourcesRequest.from_bytes(blob)
        except ValueError:
            pass


device_strategy = st.builds(
    PreparedDevice,
    type=st.sampled_from(["gpu", "partition", "vfio", "channel", "daemon"]),
    name=clean_text,
    uuid=clean_text,
    parent_uuid=clean_text,
    compute_mode=st.sampled_from(["", "SPX", "DPX", "QPX", "CPX"]),
    memory_mode=st.sampled_from(["", "NPS1", "NPS2"]),
    partition_index=st.integers(min_value=0, max_value=7),
    cdi_device_ids=st.lists(clean_text, max_size=3),
    device_nodes=st.lists(clean_text, max_size=3),
)


class TestSerdeProperties:
    @settings(max_examples=100)
    @given(dev=device_strategy)
    def test_prepared_device_round_trip(self, dev):
        d = to_dict(dev)
        json.dumps(d)  # JSON-serializable
        back = from_dict(PreparedDevice, d, strict=False)
        assert back == dev

    @settings(max_examples=50)
    @given(
        claims=st.dictionaries(
            st.uuids().map(str),
            st.builds(
                PreparedClaim,
                state=st.sampled_from(["PrepareStarted", "PrepareCompleted"]),
                claim=st.builds(ClaimRef, namespace=clean_text, name=clean_text,
                                uid=st.uuids().map(str)),
                devices=st.lists(device_strategy, max_size=3),
            ),
            max_size=4,
        )
    )
    def test_checkpoint_data_round_trip(self, claims):
        data = CheckpointData(node_boot_id="b1")
        for uid, pc in claims.items():
            data.set_claim(uid, pc)
        d = to_dict(data)
        json.dumps(d, sort_keys=True)
        back = from_dict(CheckpointData, d, strict=False)
        for uid, pc in claims.items():
            assert back.get_claim(uid) == pc

    @settings(max_examples=60)
    @given(
        ops=st.lists(
            st.tuples(
                st.sampled_from(["set", "remove", "boot"]),
                st.integers(min_value=0, max_value=7),
                st.lists(device_strategy, max_size=2),
            ),
            max_size=12,
        )
    )
    def test_canonical_payload_matches_full_serialize(self, ops):
        """The fragment-cached fast path must stay byte-identical to
        canonical JSON of the full object under ANY set/remove interleaving
        (the checksum depends on it)."""
        data = CheckpointData(node_boot_id="b0")
        for op, idx, devices in ops:
            uid = f"uid-{idx}"
            if op == "set":
                data.set_claim(uid, PreparedClaim(
                    state="PrepareCompleted",
                    claim=ClaimRef(namespace="ns", name=f"c{idx}", uid=uid),
                    devices=devices,
                ))
            elif op == "remove":
                data.remove_claim(uid)
            else:
                data.node_boot_id = f"b{idx}"
            expected = json.dumps(to_dict(data), sort_keys=True,
                                  separators=(",", ":"))
            assert data.canonical_payload() == expected

    @settings(max_examples=100)
    @given(
        n=st.integers(min_value=0, max_value=2**60),
        suffix=st.sampled_from(["", "Ki", "Mi", "Gi", "k", "M", "G"]),
    )
    def test_quantity_parse(self, n, suffix):
        mult = Quantity._SUFFIX.get(suffix, 1)
        assert Quantity.parse(f"{n}{suffix}") == n * mult


attr_name = st.from_regex(r"[a-zA-Z][a-zA-Z0-9]{0,15}", fullmatch=True)


class TestCelProperties:
    @settings(max_examples=150)
    @given(name=attr_name, val=st.text(
        alphabet=st.characters(blacklist_categories=("Cs",), blacklist_characters='"\\\\'),
        max_size=20))
    def test_string_attr_equality_total(self, name, val):
        from k8s_dra_driver_gpu_amd.k8s.celselect import CelError, cel_eval

        device = {"name": "d", "basic": {"attributes": {name: {"string": val}}}}
        expr = f'device.attributes["gpu.amd.com"].{name} == "{val}"'
        try:
            assert cel_eval(expr, "gpu.amd.com", device) is True
        except CelError:
            # reserved words / keywords may legitimately fail to parse;
            # they must never evaluate to a wrong result or escape the sandbox
            pass

    @settings(max_examples=150)
    @given(a=st.integers(-10**6, 10**6), b=st.integers(-10**6, 10**6))
    def test_int_comparisons(self, a, b):
        from k8s_dra_driver_gpu_amd.k8s.celselect import cel_eval

        device = {"name": "d", "basic": {"attributes": {"x": {"int": a}}}}
        assert cel_eval(f'device.attributes["gpu.amd.com"].x >= {b}',
                        "gpu.amd.com", device) == (a >= b)

    @settings(max_examples=100)
    @given(expr=st.text(max_size=40))
    def test_arbitrary_exprs_never_escape(self, expr):
        """Arbitrary text either evaluates to a bool or raises CelError —
        never touches the filesystem or raises anything else."""
        from k8s_dra_driver_gpu_amd.k8s.celselect import CelError, cel_eval

        device = {"name": "d", "basic": {"attributes": {}}}
        try:
            out = cel_eval(expr, "gpu.amd.com", device)
            assert isinstance(out, bool)
        except CelError:
            pass


json_scalars = st.one_of(st.none(), st.booleans(), st.integers(-10**6, 10**6),
                         st.text(max_size=12))
json_values = st.recursive(
    json_scalars,
    lambda inner: st.one_of(st.lists(inner, max_size=4),
                            st.dictionaries(st.text(max_size=10), inner, max_size=4)),
    max_leaves=12,
)


class TestConfigDecodeRobustness:
    """Opaque configs come from untrusted pod specs: the decoder + validate
    path must reject garbage with ValueError/TypeError, never crash with
    anything else (the webhook turns these into clean denials)."""

    @settings(max_examples=150)
    @given(raw=st.dictionaries(st.text(max_size=10), json_values, max_size=6))
    def test_decode_validate_never_crashes(self, raw):
        from k8s_dra_driver_gpu_amd.api.decoder import decode_config

        try:
            cfg = decode_config(raw, strict=False)
            cfg.normalize()
            cfg.validate()
        except (ValueError, TypeError):
            pass

    @settings(max_examples=150)
    @given(raw=json_values)
    def test_strict_decode_non_dict_rejected(self, raw):
        from k8s_dra_driver_gpu_amd.api.decoder import decode_config

        try:
            decode_config(raw, strict=True)
        except (ValueError, TypeError):
            pass


class TestWebhookRobustness:
    """The admission webhook receives arbitrary API-server JSON; it must
    always produce a well-formed AdmissionReview response and never raise
    (failurePolicy aside, an exception would 500 and flap the API path)."""

    @settings(max_examples=150)
    @given(review=json_values)
    def test_never_raises_always_shaped(self, review):
        from k8s_dra_driver_gpu_amd.webhook.server import validate_admission_review

        if not isinstance(review, dict):
            return
        out = validate_admission_review(review)
        assert out["kind"] == "AdmissionReview"
        assert isinstance(out["response"]["allowed"], bool)

    @settings(max_examples=100)
    @given(obj=st.dictionaries(st.text(max_size=8), json_values, max_size=4))
    def test_claim_objects_never_crash(self, obj):
        from k8s_dra_driver_gpu_amd.webhook.server import validate_admission_review

        review = {"request": {"uid": "u", "kind": {
            "group": "resource.k8s.io", "version": "v1beta1",
            "kind": "ResourceClaim"}, "object": obj}}
        out = validate_admission_review(review)
        assert isinstance(out["response"]["allowed"], bool)


class TestPartitionNameProperties:
    @settings(max_examples=100)
    @given(minor=st.integers(0, 255),
           mode=st.sampled_from(["dpx", "qpx", "cpx"]),
           idx=st.integers(0, 7))
    def test_codec_round_trip(self, minor, mode, idx):
        from k8s_dra_driver_gpu_amd.device.info import (
            format_partition_name, parse_partition_name)

        name = format_partition_name(minor, mode.upper(), idx)
        m, md, i = parse_partition_name(name)
        assert (m, md.lower(), i) == (minor, mode, idx)

    @settings(max_examples=150)
    @given(s=st.text(max_size=24))
    def test_parse_garbage_raises_cleanly(self, s):
        from k8s_dra_driver_gpu_amd.device.info import parse_partition_name

        try:
            parse_partition_name(s)
        except ValueError:
            pass


class TestCheckpointFragmentCache:
    """The checkpoint's canonical payload is composed from per-claim cached
    JSON fragments (plugin/checkpoint.py canonical_payload). The invariant:
    after ANY sequence of set/remove operations, the composed payload is
    byte-identical to serializing the whole object from scratch — a stale
    or mis-formatted fragment would silently corrupt checksums across
    driver restarts."""

    @given(ops=st.lists(
        st.tuples(st.sampled_from(["set", "remove", "reset"]),
                  st.integers(0, 7)),
        min_size=1, max_size=24))
    @settings(max_examples=60, deadline=None)
    def test_fragment_composition_matches_full_serialization(self, ops):
        from k8s_dra_driver_gpu_amd.api.serde import to_dict
        from k8s_dra_driver_gpu_amd.plugin.checkpoint import (
            CheckpointData, ClaimRef, PreparedClaim, PreparedDevice,
            _canonical,
        )

        data = CheckpointData(node_boot_id="boot-1")
        for i, (op, n) in enumerate(ops):
            uid = f"uid-{n}"
            if op == "set":
                data.set_claim(uid, PreparedClaim(
                    state="PrepareCompleted" if i % 2 else "PrepareStarted",
                    claim=ClaimRef("ns", f"c{i}", uid),
                    devices=[PreparedDevice(type="gpu", name=f"gpu-{i}")]))
            elif op == "remove":
                data.remove_claim(uid)
            else:
                # simulate a reload: fragments rebuilt lazily from raw dicts
                data = CheckpointData(
                    node_boot_id=data.node_boot_id,
                    prepared_claims=dict(data.prepared_claims))
            assert data.canonical_payload() == _canonical(to_dict(data))
