"""Phase/mode/key enums for the decentralized state machine.

Parity: /root/reference/coinstac_dinunet/config/keys.py:1-49 (Phase, Mode,
Key, AGG_Engine). The values are wire-visible (they travel in the control
JSON between sites), so they are kept stable strings.
"""


class Phase:
    INIT_RUNS = 'init_runs'
    NEXT_RUN = 'next_run'
    PRE_COMPUTATION = 'pre_computation'
    COMPUTATION = 'computation'
    NEXT_RUN_WAITING = 'next_run_waiting'
    SUCCESS = 'success'


class Mode:
    PRE_TRAIN = 'pre_train'
    TRAIN = 'train'
    VALIDATION = 'validation'
    TEST = 'test'
    VALIDATION_WAITING = 'validation_waiting'
    TRAIN_WAITING = 'train_waiting'


class Key:
    ARGS_CACHED = 'args_cached'
    TRAIN_LOG = 'train_log'
    VALIDATION_LOG = 'validation_log'
    TEST_METRICS = 'test_metrics'
    GLOBAL_TEST_METRICS = 'global_test_metrics'
    GLOBAL_TEST_SERIALIZABLE = 'global_test_serializable'
    TRAIN_SERIALIZABLE = 'train_serializable'
    VALIDATION_SERIALIZABLE = 'validation_serializable'
    TEST_SERIALIZABLE = 'test_serializable'
    DATA_CURSOR = 'data_cursor'
    DATA_LEN = 'data_len'


class AGG_Engine:
    dSGD = 'dSGD'
    powerSGD = 'powerSGD'
    rankDAD = 'rankDAD'


class GatherMode:
    # Declared for API parity with the reference (config/keys.py:47-49).
    # Like the reference, nothing consumes it yet; the engines choose their
    # collective semantics themselves (dSGD/powerSGD reduce, rankDAD gather).
    REDUCE = 'reduce'
    GATHER = 'gather'
