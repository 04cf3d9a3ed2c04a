"""CLI job entry points mirroring the reference's job mains + flags.

Reference job -> command (``python -m flink_ms_amd.cli.<name> --flags``):

  ALSImpl.scala            -> als_train
  ALSMeanVector.scala      -> als_mean_vector
  SVMImpl.scala            -> svm_train
  ALSModelGenerator.scala  -> als_model_generator
  SVMModelGenerator.scala  -> svm_model_generator
  ALSKafkaProducer.java    -> producer --model als
  SVMKafkaProducer.java    -> producer --model svm
  ALS/SVMKafkaConsumer     -> serve           (the serving job)
  ALSPredict.java          -> als_predict     (interactive)
  SVMPredict.java          -> svm_predict     (interactive)
  ALSPredictRandom.java    -> als_predict_random
  SVMPredictRandom.java    -> svm_predict_random
  RangePartitionSVMPredict -> range_partition_svm_predict
  SGD.java                 -> sgd
  MSE.java                 -> mse

Flag names/defaults follow SURVEY.md §5 (the reference's ParameterTool
inventory); Kafka/JobManager connectivity flags map onto the serving
server's host/port.
"""
