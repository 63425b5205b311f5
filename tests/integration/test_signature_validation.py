"""SignatureDef-driven Predict validation parity (VERDICT round-1 item 6).

Error message shapes must match reference predict_util.cc:66-146
character-for-character so drop-in clients see identical INVALID_ARGUMENT
details: VerifyRequestInputsSize (size mismatch with extra/missing sets),
alias-not-in-signature, output-filter alias/duplicate checks.
"""
import grpc
import pytest
import torch

from min_tfs_client_amd.server import (
    ModelServer,
    Servable,
    validate_inputs_against_signature,
    validate_output_filter,
)
from min_tfs_client_amd.turbo import TurboPredictClient

SIG = {
    "inputs": {"images": {"name": "x:0"}, "mask": {"name": "m:0"}},
    "outputs": {"probs": {"name": "p:0"}, "logits": {"name": "l:0"}},
}


def _servable():
    return Servable(
        lambda d: {"probs": torch.zeros(1), "logits": torch.zeros(1)},
        signature=SIG)


# ---------------------------------------------------------------------------
# unit: exact message shapes (predict_util.cc parity)
# ---------------------------------------------------------------------------

def test_input_size_mismatch_message():
    err = validate_inputs_against_signature(
        _servable(), {"images": 1, "mask": 1, "bogus": 1})
    assert err == (
        "input size does not match signature: 3!=2 "
        "len({bogus,images,mask}) != len({images,mask}). "
        "Sent extra: {bogus}. Missing but required: {}.")


def test_input_missing_message():
    err = validate_inputs_against_signature(_servable(), {"images": 1})
    assert err == (
        "input size does not match signature: 1!=2 "
        "len({images}) != len({images,mask}). "
        "Sent extra: {}. Missing but required: {mask}.")


def test_input_alias_not_found_message():
    # same count, wrong alias -> the per-alias branch
    err = validate_inputs_against_signature(
        _servable(), {"images": 1, "wrong": 1})
    assert err == (
        "input tensor alias not found in signature: wrong. "
        "Inputs expected to be in the set {images,mask}.")


def test_valid_inputs_pass():
    assert validate_inputs_against_signature(
        _servable(), {"images": 1, "mask": 1}) is None


def test_no_signature_accepts_anything():
    assert validate_inputs_against_signature(
        Servable(lambda d: d), {"whatever": 1}) is None


def test_output_filter_unknown_alias_message():
    err = validate_output_filter(_servable(), {}, ["nope"])
    assert err == (
        "output tensor alias not found in signature: nope "
        "Outputs expected to be in the set {logits,probs}.")


def test_output_filter_duplicate_message():
    err = validate_output_filter(_servable(), {}, ["probs", "probs"])
    assert err == "duplicate output tensor alias: probs"


# ---------------------------------------------------------------------------
# end-to-end: both server paths surface INVALID_ARGUMENT with the text
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("raw", [True, False],
                         ids=["native-raw", "proto-servicer"])
def test_end_to_end_alias_error(raw, tmp_path):
    addr = f"unix://{tmp_path}/sig_{raw}.sock" if raw \
        else "127.0.0.1:0"
    with ModelServer(address=addr, raw_predict=raw) as srv:
        srv.manager.load("sigmodel", _servable(), version=1)
        with TurboPredictClient(
                srv.address,
                backend="native" if raw else "grpcio") as client:
            with pytest.raises(grpc.RpcError) as ei:
                client.predict("sigmodel",
                               {"images": torch.zeros(2), "wrong":
                                torch.zeros(2)}, timeout=20)
            assert ei.value.code() == grpc.StatusCode.INVALID_ARGUMENT
            assert ("input tensor alias not found in signature: wrong. "
                    "Inputs expected to be in the set {images,mask}."
                    in ei.value.details())
