"""Client-level DP-FedAvgM (capability of reference
examples/dp_fed_examples/client_level_dp): unweighted noisy aggregation of
clipped deltas with adaptive clipping. Delegates to the canonical runner."""
from examples.client_dp_example.run import main

if __name__ == "__main__":
    main()
