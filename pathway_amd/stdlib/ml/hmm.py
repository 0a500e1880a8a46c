"""Hidden Markov model utilities (reference stdlib/ml/hmm behavior):
create_hmm_reducer builds a stateful reducer that tracks the most likely
current state of an HMM over a stream of observations (Viterbi update)."""
from __future__ import annotations



def create_hmm_reducer(
    graph: dict,
    beam_size: int | None = None,
    num_results_kept: int | None = None,
):
    """graph: {state: {observation: log_prob_next_states...}} encoded as
    {(state, observation): [(next_state, log_prob), ...]}; returns a
    pw.reducers.stateful_single reducer giving the most likely state."""
    import pathway_amd as pw

    @pw.reducers.stateful_single
    def hmm_state(state, observation):
        # state: dict of {hmm_state: log_prob}
        if state is None:
            state = {s: 0.0 for s, _ in graph}
        scores: dict = {}
        for (s, obs), nexts in graph.items():
            if obs != observation or s not in state:
                continue
            for ns, lp in nexts:
                cand = state[s] + lp
                if ns not in scores or cand > scores[ns]:
                    scores[ns] = cand
        if not scores:
            return state
        if beam_size is not None:
            top = sorted(scores.items(), key=lambda kv: -kv[1])[:beam_size]
            scores = dict(top)
        return scores

    return hmm_state
