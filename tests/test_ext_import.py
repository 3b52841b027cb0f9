"""The in-tree HIP extension must import on CPU-only hosts too (hipcc
cross-compiles; dlopen needs no GPU).  Catches undefined-symbol breaks
before they reach a GPU box."""


def test_extension_imports_and_has_all_ops():
    from atomo_amd.ops import ext

    e = ext()
    for fn in (
        "qsgd_pack", "qsgd_unpack_acc", "qsgd_pack_batched",
        "qsgd_unpack_batched", "svd_decode_acc", "svd_decode_batched",
        "fused_sgd", "fused_adam", "batched_gram", "batched_sel",
        "jacobi_eigh", "jacobi_eigh_big", "build_stage", "sample_stage",
    ):
        assert hasattr(e, fn), fn
