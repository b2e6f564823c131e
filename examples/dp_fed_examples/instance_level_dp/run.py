"""Instance-level DP-SGD (capability of reference
examples/dp_fed_examples/instance_level_dp). Delegates to the canonical runner."""
from examples.instance_dp_example.run import main

if __name__ == "__main__":
    main()
