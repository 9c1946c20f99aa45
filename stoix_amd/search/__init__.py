from stoix_amd.search.mcts import SearchOutput, mcts_search  # noqa: F401
