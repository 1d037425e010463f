import pytest

from realhf_amd.api.config import Abstraction, ModelInterfaceType, ModelName
from realhf_amd.api.dfg import MFCDef, build_graph


def ppo_mfcs():
    def mfc(name, role, itype, inp, out):
        return MFCDef(
            name=name,
            model_name=ModelName(role, 0),
            interface_type=itype,
            interface_impl=Abstraction("ppo_actor"),
            input_keys=tuple(inp),
            output_keys=tuple(out),
            n_seqs=128,
        )

    T = ModelInterfaceType
    return [
        mfc("actor_gen", "actor", T.GENERATE, ["packed_prompts"],
            ["packed_input_ids", "packed_logprobs", "prompt_mask"]),
        mfc("rew_inf", "reward", T.INFERENCE, ["packed_input_ids"], ["rewards"]),
        mfc("ref_inf", "ref", T.INFERENCE, ["packed_input_ids"], ["packed_ref_logprobs"]),
        mfc("critic_inf", "critic", T.INFERENCE, ["packed_input_ids"], ["values"]),
        mfc("actor_train", "actor", T.TRAIN_STEP,
            ["packed_input_ids", "packed_logprobs", "packed_ref_logprobs",
             "rewards", "values", "prompt_mask"], []),
        mfc("critic_train", "critic", T.TRAIN_STEP,
            ["packed_input_ids", "packed_logprobs", "packed_ref_logprobs",
             "rewards", "values", "prompt_mask"], []),
    ]


def test_ppo_graph_structure():
    g = build_graph(ppo_mfcs())
    gen = g.find("actor_gen")
    assert gen.is_src
    assert sorted(c.name for c in gen.children) == [
        "actor_train", "critic_inf", "critic_train", "ref_inf", "rew_inf",
    ]
    at = g.find("actor_train")
    assert at.is_dst
    assert sorted(p.name for p in at.parents) == [
        "actor_gen", "critic_inf", "ref_inf", "rew_inf",
    ]
    order = [m.name for m in g.topological_order()]
    assert order.index("actor_gen") == 0
    assert order.index("actor_train") > order.index("rew_inf")
    assert g.data_producers["rewards"].name == "rew_inf"
    assert len(g.data_consumers["packed_input_ids"]) == 5


def test_duplicate_producer_raises():
    mfcs = ppo_mfcs()
    mfcs[1].output_keys = ("packed_logprobs",)
    with pytest.raises(ValueError):
        build_graph(mfcs)


def test_roles():
    g = build_graph(ppo_mfcs())
    assert g.roles == ["actor", "critic", "ref", "reward"]
    assert [m.name for m in g.mfcs_of_role("actor")] == ["actor_gen", "actor_train"]
