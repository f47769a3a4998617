"""Fixture: tiny synthetic OpenAI-compatible LLM app factory."""


def make_app():
    from beta9_amd.integrations.llm import LLM, LLMArgs

    return LLM(args=LLMArgs(model="synthetic-llama", hidden_size=64,
                            intermediate_size=128, num_hidden_layers=2,
                            num_attention_heads=4, max_model_len=128,
                            max_num_seqs=4)).app()
