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
    # values match the reference wire/cache strings exactly
    # (config/keys.py:22-38) — they appear in out-dicts, logs.json and CSVs
    ARGS_CACHED = '_args_cached_'
    TRAIN_LOG = 'train_log'
    TRAIN_METRICS = 'train_metrics'
    TRAIN_SERIALIZABLE = 'serializable_train_scores'
    VALIDATION_LOG = 'validation_log'
    VALIDATION_METRICS = 'validation_metrics'
    VALIDATION_SERIALIZABLE = 'serializable_validation_scores'
    TEST_LOG = 'test_log'
    TEST_METRICS = 'test_metrics'
    TEST_SERIALIZABLE = 'serializable_test_scores'
    GLOBAL_TEST_LOG = 'global_test_log'
    GLOBAL_TEST_METRICS = 'global_test_metrics'
    GLOBAL_TEST_SERIALIZABLE = 'serializable_global_test_scores'
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
