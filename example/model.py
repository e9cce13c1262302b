# The example model lives in the package (tiny_deepspeed_amd/models/gpt2.py);
# re-exported here for parity with the reference's example/model.py.
from tiny_deepspeed_amd.models import GPTConfig, GPT2Model  # noqa: F401
