"""Drop-in entry matching the reference's `python3 model.py` usage
(/root/reference/README.md:12-14). The real implementation lives in the
tree_attention_torch_amd package; this shim re-exports the reference's
public surface (SURVEY.md §0 capability set) and runs the same per-GPU
spawn driver."""

from tree_attention_torch_amd import (  # noqa: F401
    TreeAttention,
    cleanup,
    flash_res_lse,
    logger,
    make_data,
    setup,
    tree_attention,
    tree_decode,
)
from tree_attention_torch_amd.main import entry, main  # noqa: F401

if __name__ == "__main__":
    entry()
