# Auto-generated R bindings (reticulate)
library(reticulate)
mmlspark_amd <- import("mmlspark_amd")

ml_access_anomaly <- function(tenantCol = NULL, indexedUserCol = NULL, indexedResCol = NULL, rankParam = NULL, regParam = NULL, maxIter = NULL, complementsetFactor = NULL, negScore = NULL, outputCol = NULL, seed = NULL) {
  stage <- mmlspark_amd$models$cyber$AccessAnomaly()
  if (!is.null(tenantCol)) stage$set("tenantCol", tenantCol)
  if (!is.null(indexedUserCol)) stage$set("indexedUserCol", indexedUserCol)
  if (!is.null(indexedResCol)) stage$set("indexedResCol", indexedResCol)
  if (!is.null(rankParam)) stage$set("rankParam", rankParam)
  if (!is.null(regParam)) stage$set("regParam", regParam)
  if (!is.null(maxIter)) stage$set("maxIter", maxIter)
  if (!is.null(complementsetFactor)) stage$set("complementsetFactor", complementsetFactor)
  if (!is.null(negScore)) stage$set("negScore", negScore)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(seed)) stage$set("seed", seed)
  stage
}

ml_access_anomaly_model <- function(tenantCol = NULL, indexedUserCol = NULL, indexedResCol = NULL, outputCol = NULL, userFactors = NULL, resFactors = NULL, scalerStats = NULL) {
  stage <- mmlspark_amd$models$cyber$AccessAnomalyModel()
  if (!is.null(tenantCol)) stage$set("tenantCol", tenantCol)
  if (!is.null(indexedUserCol)) stage$set("indexedUserCol", indexedUserCol)
  if (!is.null(indexedResCol)) stage$set("indexedResCol", indexedResCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(userFactors)) stage$set("userFactors", userFactors)
  if (!is.null(resFactors)) stage$set("resFactors", resFactors)
  if (!is.null(scalerStats)) stage$set("scalerStats", scalerStats)
  stage
}

ml_analyze_business_cards <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$AnalyzeBusinessCards()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  stage
}

ml_analyze_custom_model <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL, modelId = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$AnalyzeCustomModel()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  if (!is.null(modelId)) stage$set("modelId", modelId)
  stage
}

ml_analyze_id_documents <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$AnalyzeIDDocuments()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  stage
}

ml_analyze_image <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL, visualFeatures = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$AnalyzeImage()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  if (!is.null(visualFeatures)) stage$set("visualFeatures", visualFeatures)
  stage
}

ml_analyze_invoices <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$AnalyzeInvoices()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  stage
}

ml_analyze_layout <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$AnalyzeLayout()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  stage
}

ml_analyze_receipts <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$AnalyzeReceipts()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  stage
}

ml_azure_search_writer <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, indexDocsCol = NULL, actionType = NULL, batchSize = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$AzureSearchWriter()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(indexDocsCol)) stage$set("indexDocsCol", indexDocsCol)
  if (!is.null(actionType)) stage$set("actionType", actionType)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  stage
}

ml_best_model <- function(bestModel = NULL, allModelMetrics = NULL, bestModelMetrics = NULL, scoredDataset = NULL, rocCurve = NULL) {
  stage <- mmlspark_amd$stages$automl$BestModel()
  if (!is.null(bestModel)) stage$set("bestModel", bestModel)
  if (!is.null(allModelMetrics)) stage$set("allModelMetrics", allModelMetrics)
  if (!is.null(bestModelMetrics)) stage$set("bestModelMetrics", bestModelMetrics)
  if (!is.null(scoredDataset)) stage$set("scoredDataset", scoredDataset)
  if (!is.null(rocCurve)) stage$set("rocCurve", rocCurve)
  stage
}

ml_binary_file_reader <- function(pathCol = NULL, bytesCol = NULL) {
  stage <- mmlspark_amd$io_http$files$BinaryFileReader()
  if (!is.null(pathCol)) stage$set("pathCol", pathCol)
  if (!is.null(bytesCol)) stage$set("bytesCol", bytesCol)
  stage
}

ml_bing_image_search <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, qCol = NULL, count = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$BingImageSearch()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(qCol)) stage$set("qCol", qCol)
  if (!is.null(count)) stage$set("count", count)
  stage
}

ml_break_sentence <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, toLanguage = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$BreakSentence()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(toLanguage)) stage$set("toLanguage", toLanguage)
  stage
}

ml_cntk_model <- function(inputCol = NULL, outputCol = NULL, batchSize = NULL, moduleBytes = NULL, device = NULL, feedDict = NULL, fetchDict = NULL, convertOutputToDenseVector = NULL, batchInput = NULL, shapeOutput = NULL) {
  stage <- mmlspark_amd$models$image_featurizer$CNTKModel()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(moduleBytes)) stage$set("moduleBytes", moduleBytes)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(feedDict)) stage$set("feedDict", feedDict)
  if (!is.null(fetchDict)) stage$set("fetchDict", fetchDict)
  if (!is.null(convertOutputToDenseVector)) stage$set("convertOutputToDenseVector", convertOutputToDenseVector)
  if (!is.null(batchInput)) stage$set("batchInput", batchInput)
  if (!is.null(shapeOutput)) stage$set("shapeOutput", shapeOutput)
  stage
}

ml_cacher <- function(disable = NULL) {
  stage <- mmlspark_amd$stages$basic$Cacher()
  if (!is.null(disable)) stage$set("disable", disable)
  stage
}

ml_class_balancer <- function(inputCol = NULL, outputCol = NULL, broadcastJoin = NULL) {
  stage <- mmlspark_amd$stages$basic$ClassBalancer()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(broadcastJoin)) stage$set("broadcastJoin", broadcastJoin)
  stage
}

ml_class_balancer_model <- function(inputCol = NULL, outputCol = NULL, weights = NULL) {
  stage <- mmlspark_amd$stages$basic$ClassBalancerModel()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(weights)) stage$set("weights", weights)
  stage
}

ml_clean_missing_data <- function(inputCols = NULL, outputCols = NULL, cleaningMode = NULL, customValue = NULL) {
  stage <- mmlspark_amd$stages$featurize$CleanMissingData()
  if (!is.null(inputCols)) stage$set("inputCols", inputCols)
  if (!is.null(outputCols)) stage$set("outputCols", outputCols)
  if (!is.null(cleaningMode)) stage$set("cleaningMode", cleaningMode)
  if (!is.null(customValue)) stage$set("customValue", customValue)
  stage
}

ml_clean_missing_data_model <- function(inputCols = NULL, outputCols = NULL, fillValues = NULL) {
  stage <- mmlspark_amd$stages$featurize$CleanMissingDataModel()
  if (!is.null(inputCols)) stage$set("inputCols", inputCols)
  if (!is.null(outputCols)) stage$set("outputCols", outputCols)
  if (!is.null(fillValues)) stage$set("fillValues", fillValues)
  stage
}

ml_complement_access_transformer <- function(tenantCol = NULL, indexedUserCol = NULL, indexedResCol = NULL, complementsetFactor = NULL, seed = NULL) {
  stage <- mmlspark_amd$models$cyber$ComplementAccessTransformer()
  if (!is.null(tenantCol)) stage$set("tenantCol", tenantCol)
  if (!is.null(indexedUserCol)) stage$set("indexedUserCol", indexedUserCol)
  if (!is.null(indexedResCol)) stage$set("indexedResCol", indexedResCol)
  if (!is.null(complementsetFactor)) stage$set("complementsetFactor", complementsetFactor)
  if (!is.null(seed)) stage$set("seed", seed)
  stage
}

ml_compute_model_statistics <- function(labelCol = NULL, scoresCol = NULL, scoredLabelsCol = NULL, evaluationMetric = NULL) {
  stage <- mmlspark_amd$stages$train$ComputeModelStatistics()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(scoresCol)) stage$set("scoresCol", scoresCol)
  if (!is.null(scoredLabelsCol)) stage$set("scoredLabelsCol", scoredLabelsCol)
  if (!is.null(evaluationMetric)) stage$set("evaluationMetric", evaluationMetric)
  stage
}

ml_compute_per_instance_statistics <- function(labelCol = NULL, scoresCol = NULL, scoredLabelsCol = NULL) {
  stage <- mmlspark_amd$stages$train$ComputePerInstanceStatistics()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(scoresCol)) stage$set("scoresCol", scoresCol)
  if (!is.null(scoredLabelsCol)) stage$set("scoredLabelsCol", scoredLabelsCol)
  stage
}

ml_conditional_knn <- function(featuresCol = NULL, valuesCol = NULL, outputCol = NULL, k = NULL, batchSize = NULL, leafSize = NULL, labelCol = NULL, conditionerCol = NULL) {
  stage <- mmlspark_amd$models$knn$ConditionalKNN()
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(valuesCol)) stage$set("valuesCol", valuesCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(k)) stage$set("k", k)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(leafSize)) stage$set("leafSize", leafSize)
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(conditionerCol)) stage$set("conditionerCol", conditionerCol)
  stage
}

ml_conditional_knn_model <- function(featuresCol = NULL, valuesCol = NULL, outputCol = NULL, k = NULL, batchSize = NULL, leafSize = NULL, indexData = NULL, conditionerCol = NULL) {
  stage <- mmlspark_amd$models$knn$ConditionalKNNModel()
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(valuesCol)) stage$set("valuesCol", valuesCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(k)) stage$set("k", k)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(leafSize)) stage$set("leafSize", leafSize)
  if (!is.null(indexData)) stage$set("indexData", indexData)
  if (!is.null(conditionerCol)) stage$set("conditionerCol", conditionerCol)
  stage
}

ml_conversation_transcription <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, audioBytesCol = NULL, format = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$ConversationTranscription()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(audioBytesCol)) stage$set("audioBytesCol", audioBytesCol)
  if (!is.null(format)) stage$set("format", format)
  stage
}

ml_count_selector <- function(inputCol = NULL, outputCol = NULL) {
  stage <- mmlspark_amd$stages$featurize$CountSelector()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  stage
}

ml_count_selector_model <- function(inputCol = NULL, outputCol = NULL, indices = NULL) {
  stage <- mmlspark_amd$stages$featurize$CountSelectorModel()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(indices)) stage$set("indices", indices)
  stage
}

ml_custom_input_parser <- function(inputCol = NULL, outputCol = NULL) {
  stage <- mmlspark_amd$io_http$client$CustomInputParser()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  stage
}

ml_custom_output_parser <- function(inputCol = NULL, outputCol = NULL) {
  stage <- mmlspark_amd$io_http$client$CustomOutputParser()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  stage
}

ml_data_conversion <- function(cols = NULL, convertTo = NULL) {
  stage <- mmlspark_amd$stages$featurize$DataConversion()
  if (!is.null(cols)) stage$set("cols", cols)
  if (!is.null(convertTo)) stage$set("convertTo", convertTo)
  stage
}

ml_deep_vision_classifier <- function(labelCol = NULL, imageCol = NULL, predictionCol = NULL, modelName = NULL, imageSize = NULL, batchSize = NULL, epochs = NULL, learningRate = NULL, freezeBackbone = NULL, device = NULL) {
  stage <- mmlspark_amd$models$image_featurizer$DeepVisionClassifier()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(imageCol)) stage$set("imageCol", imageCol)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(modelName)) stage$set("modelName", modelName)
  if (!is.null(imageSize)) stage$set("imageSize", imageSize)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(epochs)) stage$set("epochs", epochs)
  if (!is.null(learningRate)) stage$set("learningRate", learningRate)
  if (!is.null(freezeBackbone)) stage$set("freezeBackbone", freezeBackbone)
  if (!is.null(device)) stage$set("device", device)
  stage
}

ml_deep_vision_model <- function(inputCol = NULL, outputCol = NULL, batchSize = NULL, moduleBytes = NULL, device = NULL, predictionCol = NULL, probabilityCol = NULL) {
  stage <- mmlspark_amd$models$image_featurizer$DeepVisionModel()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(moduleBytes)) stage$set("moduleBytes", moduleBytes)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(probabilityCol)) stage$set("probabilityCol", probabilityCol)
  stage
}

ml_describe_image <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$DescribeImage()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  stage
}

ml_detect <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, toLanguage = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$Detect()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(toLanguage)) stage$set("toLanguage", toLanguage)
  stage
}

ml_detect_entire_series <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, seriesCol = NULL, granularity = NULL, sensitivity = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$DetectEntireSeries()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(seriesCol)) stage$set("seriesCol", seriesCol)
  if (!is.null(granularity)) stage$set("granularity", granularity)
  if (!is.null(sensitivity)) stage$set("sensitivity", sensitivity)
  stage
}

ml_detect_face <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL, returnFaceAttributes = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$DetectFace()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  if (!is.null(returnFaceAttributes)) stage$set("returnFaceAttributes", returnFaceAttributes)
  stage
}

ml_detect_last_anomaly <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, seriesCol = NULL, granularity = NULL, sensitivity = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$DetectLastAnomaly()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(seriesCol)) stage$set("seriesCol", seriesCol)
  if (!is.null(granularity)) stage$set("granularity", granularity)
  if (!is.null(sensitivity)) stage$set("sensitivity", sensitivity)
  stage
}

ml_dictionary_examples <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, toLanguage = NULL, translationCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$DictionaryExamples()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(toLanguage)) stage$set("toLanguage", toLanguage)
  if (!is.null(translationCol)) stage$set("translationCol", translationCol)
  stage
}

ml_dictionary_lookup <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, toLanguage = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$DictionaryLookup()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(toLanguage)) stage$set("toLanguage", toLanguage)
  stage
}

ml_document_translator <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, sourceUrlCol = NULL, targetUrlCol = NULL, targetLanguage = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$DocumentTranslator()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(sourceUrlCol)) stage$set("sourceUrlCol", sourceUrlCol)
  if (!is.null(targetUrlCol)) stage$set("targetUrlCol", targetUrlCol)
  if (!is.null(targetLanguage)) stage$set("targetLanguage", targetLanguage)
  stage
}

ml_drop_columns <- function(cols = NULL) {
  stage <- mmlspark_amd$stages$basic$DropColumns()
  if (!is.null(cols)) stage$set("cols", cols)
  stage
}

ml_drop_http_errors <- function(inputCol = NULL, errorCol = NULL) {
  stage <- mmlspark_amd$io_http$client$DropHTTPErrors()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  stage
}

ml_dynamic_mini_batch_transformer <- function(maxBatchSize = NULL) {
  stage <- mmlspark_amd$stages$batching$DynamicMiniBatchTransformer()
  if (!is.null(maxBatchSize)) stage$set("maxBatchSize", maxBatchSize)
  stage
}

ml_ensemble_by_key <- function(keys = NULL, cols = NULL, strategy = NULL, collapseGroup = NULL) {
  stage <- mmlspark_amd$stages$basic$EnsembleByKey()
  if (!is.null(keys)) stage$set("keys", keys)
  if (!is.null(cols)) stage$set("cols", cols)
  if (!is.null(strategy)) stage$set("strategy", strategy)
  if (!is.null(collapseGroup)) stage$set("collapseGroup", collapseGroup)
  stage
}

ml_entity_detector <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, language = NULL, languageCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$EntityDetector()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(language)) stage$set("language", language)
  if (!is.null(languageCol)) stage$set("languageCol", languageCol)
  stage
}

ml_entity_detector_v2 <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, language = NULL, languageCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$EntityDetectorV2()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(language)) stage$set("language", language)
  if (!is.null(languageCol)) stage$set("languageCol", languageCol)
  stage
}

ml_explode <- function(inputCol = NULL, outputCol = NULL) {
  stage <- mmlspark_amd$stages$basic$Explode()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  stage
}

ml_fast_vector_assembler <- function(inputCols = NULL, outputCol = NULL) {
  stage <- mmlspark_amd$stages$featurize$FastVectorAssembler()
  if (!is.null(inputCols)) stage$set("inputCols", inputCols)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  stage
}

ml_featurize <- function(inputCols = NULL, outputCol = NULL, oneHotEncodeCategoricals = NULL, numFeatures = NULL) {
  stage <- mmlspark_amd$stages$featurize$Featurize()
  if (!is.null(inputCols)) stage$set("inputCols", inputCols)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(oneHotEncodeCategoricals)) stage$set("oneHotEncodeCategoricals", oneHotEncodeCategoricals)
  if (!is.null(numFeatures)) stage$set("numFeatures", numFeatures)
  stage
}

ml_featurize_model <- function(plan = NULL, outputCol = NULL) {
  stage <- mmlspark_amd$stages$featurize$FeaturizeModel()
  if (!is.null(plan)) stage$set("plan", plan)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  stage
}

ml_find_best_model <- function(evaluationMetric = NULL, labelCol = NULL, models = NULL) {
  stage <- mmlspark_amd$stages$automl$FindBestModel()
  if (!is.null(evaluationMetric)) stage$set("evaluationMetric", evaluationMetric)
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(models)) stage$set("models", models)
  stage
}

ml_find_similar_face <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, faceIdCol = NULL, faceIdsCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$FindSimilarFace()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(faceIdCol)) stage$set("faceIdCol", faceIdCol)
  if (!is.null(faceIdsCol)) stage$set("faceIdsCol", faceIdsCol)
  stage
}

ml_fixed_mini_batch_transformer <- function(batchSize = NULL, maxBufferSize = NULL) {
  stage <- mmlspark_amd$stages$batching$FixedMiniBatchTransformer()
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(maxBufferSize)) stage$set("maxBufferSize", maxBufferSize)
  stage
}

ml_flatten_batch <- function() {
  stage <- mmlspark_amd$stages$batching$FlattenBatch()

  stage
}

ml_generate_thumbnails <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL, width = NULL, height = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$GenerateThumbnails()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  if (!is.null(width)) stage$set("width", width)
  if (!is.null(height)) stage$set("height", height)
  stage
}

ml_get_custom_model <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL, modelId = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$GetCustomModel()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  if (!is.null(modelId)) stage$set("modelId", modelId)
  stage
}

ml_group_faces <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, faceIdsCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$GroupFaces()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(faceIdsCol)) stage$set("faceIdsCol", faceIdsCol)
  stage
}

ml_http_transformer <- function(inputCol = NULL, outputCol = NULL, concurrency = NULL, timeout = NULL) {
  stage <- mmlspark_amd$io_http$client$HTTPTransformer()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  stage
}

ml_id_indexer <- function(inputCol = NULL, partitionKey = NULL, outputCol = NULL, resetPerPartition = NULL) {
  stage <- mmlspark_amd$models$cyber$IdIndexer()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(partitionKey)) stage$set("partitionKey", partitionKey)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(resetPerPartition)) stage$set("resetPerPartition", resetPerPartition)
  stage
}

ml_id_indexer_model <- function(inputCol = NULL, partitionKey = NULL, outputCol = NULL, resetPerPartition = NULL, idMaps = NULL) {
  stage <- mmlspark_amd$models$cyber$IdIndexerModel()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(partitionKey)) stage$set("partitionKey", partitionKey)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(resetPerPartition)) stage$set("resetPerPartition", resetPerPartition)
  if (!is.null(idMaps)) stage$set("idMaps", idMaps)
  stage
}

ml_identify_faces <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, faceIdsCol = NULL, personGroupId = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$IdentifyFaces()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(faceIdsCol)) stage$set("faceIdsCol", faceIdsCol)
  if (!is.null(personGroupId)) stage$set("personGroupId", personGroupId)
  stage
}

ml_image_featurizer <- function(inputCol = NULL, outputCol = NULL, batchSize = NULL, moduleBytes = NULL, device = NULL, modelName = NULL, cutOutputLayers = NULL, imageSize = NULL, modelPath = NULL) {
  stage <- mmlspark_amd$models$image_featurizer$ImageFeaturizer()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(moduleBytes)) stage$set("moduleBytes", moduleBytes)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(modelName)) stage$set("modelName", modelName)
  if (!is.null(cutOutputLayers)) stage$set("cutOutputLayers", cutOutputLayers)
  if (!is.null(imageSize)) stage$set("imageSize", imageSize)
  if (!is.null(modelPath)) stage$set("modelPath", modelPath)
  stage
}

ml_image_lime <- function(model = NULL, targetCol = NULL, targetClasses = NULL, outputCol = NULL, numSamples = NULL, metricsCol = NULL, seed = NULL, rowBatch = NULL, kernelWidth = NULL, regularization = NULL, inputCol = NULL, cellSize = NULL, modifier = NULL, superpixelCol = NULL) {
  stage <- mmlspark_amd$explainers$lime$ImageLIME()
  if (!is.null(model)) stage$set("model", model)
  if (!is.null(targetCol)) stage$set("targetCol", targetCol)
  if (!is.null(targetClasses)) stage$set("targetClasses", targetClasses)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(numSamples)) stage$set("numSamples", numSamples)
  if (!is.null(metricsCol)) stage$set("metricsCol", metricsCol)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(rowBatch)) stage$set("rowBatch", rowBatch)
  if (!is.null(kernelWidth)) stage$set("kernelWidth", kernelWidth)
  if (!is.null(regularization)) stage$set("regularization", regularization)
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(cellSize)) stage$set("cellSize", cellSize)
  if (!is.null(modifier)) stage$set("modifier", modifier)
  if (!is.null(superpixelCol)) stage$set("superpixelCol", superpixelCol)
  stage
}

ml_image_reader <- function(bytesCol = NULL, imageCol = NULL, dropInvalid = NULL) {
  stage <- mmlspark_amd$io_http$files$ImageReader()
  if (!is.null(bytesCol)) stage$set("bytesCol", bytesCol)
  if (!is.null(imageCol)) stage$set("imageCol", imageCol)
  if (!is.null(dropInvalid)) stage$set("dropInvalid", dropInvalid)
  stage
}

ml_image_shap <- function(model = NULL, targetCol = NULL, targetClasses = NULL, outputCol = NULL, numSamples = NULL, metricsCol = NULL, seed = NULL, rowBatch = NULL, inputCol = NULL, cellSize = NULL, modifier = NULL, superpixelCol = NULL) {
  stage <- mmlspark_amd$explainers$shap$ImageSHAP()
  if (!is.null(model)) stage$set("model", model)
  if (!is.null(targetCol)) stage$set("targetCol", targetCol)
  if (!is.null(targetClasses)) stage$set("targetClasses", targetClasses)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(numSamples)) stage$set("numSamples", numSamples)
  if (!is.null(metricsCol)) stage$set("metricsCol", metricsCol)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(rowBatch)) stage$set("rowBatch", rowBatch)
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(cellSize)) stage$set("cellSize", cellSize)
  if (!is.null(modifier)) stage$set("modifier", modifier)
  if (!is.null(superpixelCol)) stage$set("superpixelCol", superpixelCol)
  stage
}

ml_image_set_augmenter <- function(inputCol = NULL, outputCol = NULL, flipLeftRight = NULL, flipUpDown = NULL) {
  stage <- mmlspark_amd$models$images$ImageSetAugmenter()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(flipLeftRight)) stage$set("flipLeftRight", flipLeftRight)
  if (!is.null(flipUpDown)) stage$set("flipUpDown", flipUpDown)
  stage
}

ml_image_transformer <- function(inputCol = NULL, outputCol = NULL, stages = NULL, outputType = NULL) {
  stage <- mmlspark_amd$models$images$ImageTransformer()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(stages)) stage$set("stages", stages)
  if (!is.null(outputType)) stage$set("outputType", outputType)
  stage
}

ml_index_to_value <- function(inputCol = NULL, outputCol = NULL, levels = NULL) {
  stage <- mmlspark_amd$stages$featurize$IndexToValue()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(levels)) stage$set("levels", levels)
  stage
}

ml_isolation_forest <- function(featuresCol = NULL, featureCols = NULL, predictionCol = NULL, scoreCol = NULL, numEstimators = NULL, maxSamples = NULL, maxFeatures = NULL, bootstrap = NULL, contamination = NULL, randomSeed = NULL) {
  stage <- mmlspark_amd$models$iforest$IsolationForest()
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(featureCols)) stage$set("featureCols", featureCols)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(scoreCol)) stage$set("scoreCol", scoreCol)
  if (!is.null(numEstimators)) stage$set("numEstimators", numEstimators)
  if (!is.null(maxSamples)) stage$set("maxSamples", maxSamples)
  if (!is.null(maxFeatures)) stage$set("maxFeatures", maxFeatures)
  if (!is.null(bootstrap)) stage$set("bootstrap", bootstrap)
  if (!is.null(contamination)) stage$set("contamination", contamination)
  if (!is.null(randomSeed)) stage$set("randomSeed", randomSeed)
  stage
}

ml_isolation_forest_model <- function(featuresCol = NULL, featureCols = NULL, predictionCol = NULL, scoreCol = NULL, scoreThreshold = NULL, forestArrays = NULL) {
  stage <- mmlspark_amd$models$iforest$IsolationForestModel()
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(featureCols)) stage$set("featureCols", featureCols)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(scoreCol)) stage$set("scoreCol", scoreCol)
  if (!is.null(scoreThreshold)) stage$set("scoreThreshold", scoreThreshold)
  if (!is.null(forestArrays)) stage$set("forestArrays", forestArrays)
  stage
}

ml_json_input_parser <- function(inputCol = NULL, outputCol = NULL, url = NULL, method = NULL, headers = NULL) {
  stage <- mmlspark_amd$io_http$client$JSONInputParser()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(method)) stage$set("method", method)
  if (!is.null(headers)) stage$set("headers", headers)
  stage
}

ml_json_output_parser <- function(inputCol = NULL, outputCol = NULL) {
  stage <- mmlspark_amd$io_http$client$JSONOutputParser()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  stage
}

ml_knn <- function(featuresCol = NULL, valuesCol = NULL, outputCol = NULL, k = NULL, batchSize = NULL, leafSize = NULL) {
  stage <- mmlspark_amd$models$knn$KNN()
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(valuesCol)) stage$set("valuesCol", valuesCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(k)) stage$set("k", k)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(leafSize)) stage$set("leafSize", leafSize)
  stage
}

ml_knn_model <- function(featuresCol = NULL, valuesCol = NULL, outputCol = NULL, k = NULL, batchSize = NULL, leafSize = NULL, indexData = NULL) {
  stage <- mmlspark_amd$models$knn$KNNModel()
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(valuesCol)) stage$set("valuesCol", valuesCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(k)) stage$set("k", k)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(leafSize)) stage$set("leafSize", leafSize)
  if (!is.null(indexData)) stage$set("indexData", indexData)
  stage
}

ml_key_phrase_extractor <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, language = NULL, languageCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$KeyPhraseExtractor()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(language)) stage$set("language", language)
  if (!is.null(languageCol)) stage$set("languageCol", languageCol)
  stage
}

ml_key_phrase_extractor_v2 <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, language = NULL, languageCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$KeyPhraseExtractorV2()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(language)) stage$set("language", language)
  if (!is.null(languageCol)) stage$set("languageCol", languageCol)
  stage
}

ml_lambda <- function() {
  stage <- mmlspark_amd$stages$basic$Lambda()

  stage
}

ml_language_detector <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, language = NULL, languageCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$LanguageDetector()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(language)) stage$set("language", language)
  if (!is.null(languageCol)) stage$set("languageCol", languageCol)
  stage
}

ml_language_detector_v2 <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, language = NULL, languageCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$LanguageDetectorV2()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(language)) stage$set("language", language)
  if (!is.null(languageCol)) stage$set("languageCol", languageCol)
  stage
}

ml_light_gbm_classification_model <- function(labelCol = NULL, featuresCol = NULL, featureCols = NULL, predictionCol = NULL, leafPredictionCol = NULL, featuresShapCol = NULL, boosterModelStr = NULL, startIteration = NULL, numIterations = NULL, device = NULL, rawPredictionCol = NULL, probabilityCol = NULL, thresholds = NULL, isUnbalance = NULL) {
  stage <- mmlspark_amd$models$gbdt$estimators$LightGBMClassificationModel()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(featureCols)) stage$set("featureCols", featureCols)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(leafPredictionCol)) stage$set("leafPredictionCol", leafPredictionCol)
  if (!is.null(featuresShapCol)) stage$set("featuresShapCol", featuresShapCol)
  if (!is.null(boosterModelStr)) stage$set("boosterModelStr", boosterModelStr)
  if (!is.null(startIteration)) stage$set("startIteration", startIteration)
  if (!is.null(numIterations)) stage$set("numIterations", numIterations)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(rawPredictionCol)) stage$set("rawPredictionCol", rawPredictionCol)
  if (!is.null(probabilityCol)) stage$set("probabilityCol", probabilityCol)
  if (!is.null(thresholds)) stage$set("thresholds", thresholds)
  if (!is.null(isUnbalance)) stage$set("isUnbalance", isUnbalance)
  stage
}

ml_light_gbm_classifier <- function(labelCol = NULL, featuresCol = NULL, featureCols = NULL, weightCol = NULL, validationIndicatorCol = NULL, initScoreCol = NULL, predictionCol = NULL, numIterations = NULL, learningRate = NULL, numLeaves = NULL, maxDepth = NULL, maxBin = NULL, lambdaL1 = NULL, lambdaL2 = NULL, minDataInLeaf = NULL, minSumHessianInLeaf = NULL, minGainToSplit = NULL, featureFraction = NULL, baggingFraction = NULL, baggingFreq = NULL, baggingSeed = NULL, boostingType = NULL, topRate = NULL, otherRate = NULL, dropRate = NULL, skipDrop = NULL, maxDrop = NULL, maxDeltaStep = NULL, earlyStoppingRound = NULL, objective = NULL, metric = NULL, seed = NULL, numBatches = NULL, verbosity = NULL, isProvideTrainingMetric = NULL, useBarrierExecutionMode = NULL, parallelism = NULL, topK = NULL, categoricalSlotIndexes = NULL, categoricalSlotNames = NULL, slotNames = NULL, modelString = NULL, lightGBMBooster = NULL, fobj = NULL, boostFromAverage = NULL, improvementTolerance = NULL, posBaggingFraction = NULL, negBaggingFraction = NULL, binSampleCount = NULL, maxBinByFeature = NULL, uniformDrop = NULL, xgboostDartMode = NULL, startIteration = NULL, leafPredictionCol = NULL, featuresShapCol = NULL, chunkSize = NULL, defaultListenPort = NULL, driverListenPort = NULL, timeout = NULL, numTasks = NULL, numThreads = NULL, useSingleDatasetMode = NULL, matrixType = NULL, checkpointDir = NULL, checkpointInterval = NULL, device = NULL, rawPredictionCol = NULL, probabilityCol = NULL, isUnbalance = NULL) {
  stage <- mmlspark_amd$models$gbdt$estimators$LightGBMClassifier()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(featureCols)) stage$set("featureCols", featureCols)
  if (!is.null(weightCol)) stage$set("weightCol", weightCol)
  if (!is.null(validationIndicatorCol)) stage$set("validationIndicatorCol", validationIndicatorCol)
  if (!is.null(initScoreCol)) stage$set("initScoreCol", initScoreCol)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(numIterations)) stage$set("numIterations", numIterations)
  if (!is.null(learningRate)) stage$set("learningRate", learningRate)
  if (!is.null(numLeaves)) stage$set("numLeaves", numLeaves)
  if (!is.null(maxDepth)) stage$set("maxDepth", maxDepth)
  if (!is.null(maxBin)) stage$set("maxBin", maxBin)
  if (!is.null(lambdaL1)) stage$set("lambdaL1", lambdaL1)
  if (!is.null(lambdaL2)) stage$set("lambdaL2", lambdaL2)
  if (!is.null(minDataInLeaf)) stage$set("minDataInLeaf", minDataInLeaf)
  if (!is.null(minSumHessianInLeaf)) stage$set("minSumHessianInLeaf", minSumHessianInLeaf)
  if (!is.null(minGainToSplit)) stage$set("minGainToSplit", minGainToSplit)
  if (!is.null(featureFraction)) stage$set("featureFraction", featureFraction)
  if (!is.null(baggingFraction)) stage$set("baggingFraction", baggingFraction)
  if (!is.null(baggingFreq)) stage$set("baggingFreq", baggingFreq)
  if (!is.null(baggingSeed)) stage$set("baggingSeed", baggingSeed)
  if (!is.null(boostingType)) stage$set("boostingType", boostingType)
  if (!is.null(topRate)) stage$set("topRate", topRate)
  if (!is.null(otherRate)) stage$set("otherRate", otherRate)
  if (!is.null(dropRate)) stage$set("dropRate", dropRate)
  if (!is.null(skipDrop)) stage$set("skipDrop", skipDrop)
  if (!is.null(maxDrop)) stage$set("maxDrop", maxDrop)
  if (!is.null(maxDeltaStep)) stage$set("maxDeltaStep", maxDeltaStep)
  if (!is.null(earlyStoppingRound)) stage$set("earlyStoppingRound", earlyStoppingRound)
  if (!is.null(objective)) stage$set("objective", objective)
  if (!is.null(metric)) stage$set("metric", metric)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(numBatches)) stage$set("numBatches", numBatches)
  if (!is.null(verbosity)) stage$set("verbosity", verbosity)
  if (!is.null(isProvideTrainingMetric)) stage$set("isProvideTrainingMetric", isProvideTrainingMetric)
  if (!is.null(useBarrierExecutionMode)) stage$set("useBarrierExecutionMode", useBarrierExecutionMode)
  if (!is.null(parallelism)) stage$set("parallelism", parallelism)
  if (!is.null(topK)) stage$set("topK", topK)
  if (!is.null(categoricalSlotIndexes)) stage$set("categoricalSlotIndexes", categoricalSlotIndexes)
  if (!is.null(categoricalSlotNames)) stage$set("categoricalSlotNames", categoricalSlotNames)
  if (!is.null(slotNames)) stage$set("slotNames", slotNames)
  if (!is.null(modelString)) stage$set("modelString", modelString)
  if (!is.null(lightGBMBooster)) stage$set("lightGBMBooster", lightGBMBooster)
  if (!is.null(fobj)) stage$set("fobj", fobj)
  if (!is.null(boostFromAverage)) stage$set("boostFromAverage", boostFromAverage)
  if (!is.null(improvementTolerance)) stage$set("improvementTolerance", improvementTolerance)
  if (!is.null(posBaggingFraction)) stage$set("posBaggingFraction", posBaggingFraction)
  if (!is.null(negBaggingFraction)) stage$set("negBaggingFraction", negBaggingFraction)
  if (!is.null(binSampleCount)) stage$set("binSampleCount", binSampleCount)
  if (!is.null(maxBinByFeature)) stage$set("maxBinByFeature", maxBinByFeature)
  if (!is.null(uniformDrop)) stage$set("uniformDrop", uniformDrop)
  if (!is.null(xgboostDartMode)) stage$set("xgboostDartMode", xgboostDartMode)
  if (!is.null(startIteration)) stage$set("startIteration", startIteration)
  if (!is.null(leafPredictionCol)) stage$set("leafPredictionCol", leafPredictionCol)
  if (!is.null(featuresShapCol)) stage$set("featuresShapCol", featuresShapCol)
  if (!is.null(chunkSize)) stage$set("chunkSize", chunkSize)
  if (!is.null(defaultListenPort)) stage$set("defaultListenPort", defaultListenPort)
  if (!is.null(driverListenPort)) stage$set("driverListenPort", driverListenPort)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(numTasks)) stage$set("numTasks", numTasks)
  if (!is.null(numThreads)) stage$set("numThreads", numThreads)
  if (!is.null(useSingleDatasetMode)) stage$set("useSingleDatasetMode", useSingleDatasetMode)
  if (!is.null(matrixType)) stage$set("matrixType", matrixType)
  if (!is.null(checkpointDir)) stage$set("checkpointDir", checkpointDir)
  if (!is.null(checkpointInterval)) stage$set("checkpointInterval", checkpointInterval)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(rawPredictionCol)) stage$set("rawPredictionCol", rawPredictionCol)
  if (!is.null(probabilityCol)) stage$set("probabilityCol", probabilityCol)
  if (!is.null(isUnbalance)) stage$set("isUnbalance", isUnbalance)
  stage
}

ml_light_gbm_ranker <- function(labelCol = NULL, featuresCol = NULL, featureCols = NULL, weightCol = NULL, validationIndicatorCol = NULL, initScoreCol = NULL, predictionCol = NULL, numIterations = NULL, learningRate = NULL, numLeaves = NULL, maxDepth = NULL, maxBin = NULL, lambdaL1 = NULL, lambdaL2 = NULL, minDataInLeaf = NULL, minSumHessianInLeaf = NULL, minGainToSplit = NULL, featureFraction = NULL, baggingFraction = NULL, baggingFreq = NULL, baggingSeed = NULL, boostingType = NULL, topRate = NULL, otherRate = NULL, dropRate = NULL, skipDrop = NULL, maxDrop = NULL, maxDeltaStep = NULL, earlyStoppingRound = NULL, objective = NULL, metric = NULL, seed = NULL, numBatches = NULL, verbosity = NULL, isProvideTrainingMetric = NULL, useBarrierExecutionMode = NULL, parallelism = NULL, topK = NULL, categoricalSlotIndexes = NULL, categoricalSlotNames = NULL, slotNames = NULL, modelString = NULL, lightGBMBooster = NULL, fobj = NULL, boostFromAverage = NULL, improvementTolerance = NULL, posBaggingFraction = NULL, negBaggingFraction = NULL, binSampleCount = NULL, maxBinByFeature = NULL, uniformDrop = NULL, xgboostDartMode = NULL, startIteration = NULL, leafPredictionCol = NULL, featuresShapCol = NULL, chunkSize = NULL, defaultListenPort = NULL, driverListenPort = NULL, timeout = NULL, numTasks = NULL, numThreads = NULL, useSingleDatasetMode = NULL, matrixType = NULL, checkpointDir = NULL, checkpointInterval = NULL, device = NULL, groupCol = NULL, labelGain = NULL, maxPosition = NULL, evalAt = NULL, repartitionByGroupingColumn = NULL) {
  stage <- mmlspark_amd$models$gbdt$estimators$LightGBMRanker()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(featureCols)) stage$set("featureCols", featureCols)
  if (!is.null(weightCol)) stage$set("weightCol", weightCol)
  if (!is.null(validationIndicatorCol)) stage$set("validationIndicatorCol", validationIndicatorCol)
  if (!is.null(initScoreCol)) stage$set("initScoreCol", initScoreCol)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(numIterations)) stage$set("numIterations", numIterations)
  if (!is.null(learningRate)) stage$set("learningRate", learningRate)
  if (!is.null(numLeaves)) stage$set("numLeaves", numLeaves)
  if (!is.null(maxDepth)) stage$set("maxDepth", maxDepth)
  if (!is.null(maxBin)) stage$set("maxBin", maxBin)
  if (!is.null(lambdaL1)) stage$set("lambdaL1", lambdaL1)
  if (!is.null(lambdaL2)) stage$set("lambdaL2", lambdaL2)
  if (!is.null(minDataInLeaf)) stage$set("minDataInLeaf", minDataInLeaf)
  if (!is.null(minSumHessianInLeaf)) stage$set("minSumHessianInLeaf", minSumHessianInLeaf)
  if (!is.null(minGainToSplit)) stage$set("minGainToSplit", minGainToSplit)
  if (!is.null(featureFraction)) stage$set("featureFraction", featureFraction)
  if (!is.null(baggingFraction)) stage$set("baggingFraction", baggingFraction)
  if (!is.null(baggingFreq)) stage$set("baggingFreq", baggingFreq)
  if (!is.null(baggingSeed)) stage$set("baggingSeed", baggingSeed)
  if (!is.null(boostingType)) stage$set("boostingType", boostingType)
  if (!is.null(topRate)) stage$set("topRate", topRate)
  if (!is.null(otherRate)) stage$set("otherRate", otherRate)
  if (!is.null(dropRate)) stage$set("dropRate", dropRate)
  if (!is.null(skipDrop)) stage$set("skipDrop", skipDrop)
  if (!is.null(maxDrop)) stage$set("maxDrop", maxDrop)
  if (!is.null(maxDeltaStep)) stage$set("maxDeltaStep", maxDeltaStep)
  if (!is.null(earlyStoppingRound)) stage$set("earlyStoppingRound", earlyStoppingRound)
  if (!is.null(objective)) stage$set("objective", objective)
  if (!is.null(metric)) stage$set("metric", metric)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(numBatches)) stage$set("numBatches", numBatches)
  if (!is.null(verbosity)) stage$set("verbosity", verbosity)
  if (!is.null(isProvideTrainingMetric)) stage$set("isProvideTrainingMetric", isProvideTrainingMetric)
  if (!is.null(useBarrierExecutionMode)) stage$set("useBarrierExecutionMode", useBarrierExecutionMode)
  if (!is.null(parallelism)) stage$set("parallelism", parallelism)
  if (!is.null(topK)) stage$set("topK", topK)
  if (!is.null(categoricalSlotIndexes)) stage$set("categoricalSlotIndexes", categoricalSlotIndexes)
  if (!is.null(categoricalSlotNames)) stage$set("categoricalSlotNames", categoricalSlotNames)
  if (!is.null(slotNames)) stage$set("slotNames", slotNames)
  if (!is.null(modelString)) stage$set("modelString", modelString)
  if (!is.null(lightGBMBooster)) stage$set("lightGBMBooster", lightGBMBooster)
  if (!is.null(fobj)) stage$set("fobj", fobj)
  if (!is.null(boostFromAverage)) stage$set("boostFromAverage", boostFromAverage)
  if (!is.null(improvementTolerance)) stage$set("improvementTolerance", improvementTolerance)
  if (!is.null(posBaggingFraction)) stage$set("posBaggingFraction", posBaggingFraction)
  if (!is.null(negBaggingFraction)) stage$set("negBaggingFraction", negBaggingFraction)
  if (!is.null(binSampleCount)) stage$set("binSampleCount", binSampleCount)
  if (!is.null(maxBinByFeature)) stage$set("maxBinByFeature", maxBinByFeature)
  if (!is.null(uniformDrop)) stage$set("uniformDrop", uniformDrop)
  if (!is.null(xgboostDartMode)) stage$set("xgboostDartMode", xgboostDartMode)
  if (!is.null(startIteration)) stage$set("startIteration", startIteration)
  if (!is.null(leafPredictionCol)) stage$set("leafPredictionCol", leafPredictionCol)
  if (!is.null(featuresShapCol)) stage$set("featuresShapCol", featuresShapCol)
  if (!is.null(chunkSize)) stage$set("chunkSize", chunkSize)
  if (!is.null(defaultListenPort)) stage$set("defaultListenPort", defaultListenPort)
  if (!is.null(driverListenPort)) stage$set("driverListenPort", driverListenPort)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(numTasks)) stage$set("numTasks", numTasks)
  if (!is.null(numThreads)) stage$set("numThreads", numThreads)
  if (!is.null(useSingleDatasetMode)) stage$set("useSingleDatasetMode", useSingleDatasetMode)
  if (!is.null(matrixType)) stage$set("matrixType", matrixType)
  if (!is.null(checkpointDir)) stage$set("checkpointDir", checkpointDir)
  if (!is.null(checkpointInterval)) stage$set("checkpointInterval", checkpointInterval)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(groupCol)) stage$set("groupCol", groupCol)
  if (!is.null(labelGain)) stage$set("labelGain", labelGain)
  if (!is.null(maxPosition)) stage$set("maxPosition", maxPosition)
  if (!is.null(evalAt)) stage$set("evalAt", evalAt)
  if (!is.null(repartitionByGroupingColumn)) stage$set("repartitionByGroupingColumn", repartitionByGroupingColumn)
  stage
}

ml_light_gbm_ranker_model <- function(labelCol = NULL, featuresCol = NULL, featureCols = NULL, predictionCol = NULL, leafPredictionCol = NULL, featuresShapCol = NULL, boosterModelStr = NULL, startIteration = NULL, numIterations = NULL, device = NULL, labelGain = NULL, maxPosition = NULL, evalAt = NULL) {
  stage <- mmlspark_amd$models$gbdt$estimators$LightGBMRankerModel()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(featureCols)) stage$set("featureCols", featureCols)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(leafPredictionCol)) stage$set("leafPredictionCol", leafPredictionCol)
  if (!is.null(featuresShapCol)) stage$set("featuresShapCol", featuresShapCol)
  if (!is.null(boosterModelStr)) stage$set("boosterModelStr", boosterModelStr)
  if (!is.null(startIteration)) stage$set("startIteration", startIteration)
  if (!is.null(numIterations)) stage$set("numIterations", numIterations)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(labelGain)) stage$set("labelGain", labelGain)
  if (!is.null(maxPosition)) stage$set("maxPosition", maxPosition)
  if (!is.null(evalAt)) stage$set("evalAt", evalAt)
  stage
}

ml_light_gbm_regression_model <- function(labelCol = NULL, featuresCol = NULL, featureCols = NULL, predictionCol = NULL, leafPredictionCol = NULL, featuresShapCol = NULL, boosterModelStr = NULL, startIteration = NULL, numIterations = NULL, device = NULL, alpha = NULL, tweedieVariancePower = NULL) {
  stage <- mmlspark_amd$models$gbdt$estimators$LightGBMRegressionModel()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(featureCols)) stage$set("featureCols", featureCols)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(leafPredictionCol)) stage$set("leafPredictionCol", leafPredictionCol)
  if (!is.null(featuresShapCol)) stage$set("featuresShapCol", featuresShapCol)
  if (!is.null(boosterModelStr)) stage$set("boosterModelStr", boosterModelStr)
  if (!is.null(startIteration)) stage$set("startIteration", startIteration)
  if (!is.null(numIterations)) stage$set("numIterations", numIterations)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(alpha)) stage$set("alpha", alpha)
  if (!is.null(tweedieVariancePower)) stage$set("tweedieVariancePower", tweedieVariancePower)
  stage
}

ml_light_gbm_regressor <- function(labelCol = NULL, featuresCol = NULL, featureCols = NULL, weightCol = NULL, validationIndicatorCol = NULL, initScoreCol = NULL, predictionCol = NULL, numIterations = NULL, learningRate = NULL, numLeaves = NULL, maxDepth = NULL, maxBin = NULL, lambdaL1 = NULL, lambdaL2 = NULL, minDataInLeaf = NULL, minSumHessianInLeaf = NULL, minGainToSplit = NULL, featureFraction = NULL, baggingFraction = NULL, baggingFreq = NULL, baggingSeed = NULL, boostingType = NULL, topRate = NULL, otherRate = NULL, dropRate = NULL, skipDrop = NULL, maxDrop = NULL, maxDeltaStep = NULL, earlyStoppingRound = NULL, objective = NULL, metric = NULL, seed = NULL, numBatches = NULL, verbosity = NULL, isProvideTrainingMetric = NULL, useBarrierExecutionMode = NULL, parallelism = NULL, topK = NULL, categoricalSlotIndexes = NULL, categoricalSlotNames = NULL, slotNames = NULL, modelString = NULL, lightGBMBooster = NULL, fobj = NULL, boostFromAverage = NULL, improvementTolerance = NULL, posBaggingFraction = NULL, negBaggingFraction = NULL, binSampleCount = NULL, maxBinByFeature = NULL, uniformDrop = NULL, xgboostDartMode = NULL, startIteration = NULL, leafPredictionCol = NULL, featuresShapCol = NULL, chunkSize = NULL, defaultListenPort = NULL, driverListenPort = NULL, timeout = NULL, numTasks = NULL, numThreads = NULL, useSingleDatasetMode = NULL, matrixType = NULL, checkpointDir = NULL, checkpointInterval = NULL, device = NULL, alpha = NULL, tweedieVariancePower = NULL) {
  stage <- mmlspark_amd$models$gbdt$estimators$LightGBMRegressor()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(featureCols)) stage$set("featureCols", featureCols)
  if (!is.null(weightCol)) stage$set("weightCol", weightCol)
  if (!is.null(validationIndicatorCol)) stage$set("validationIndicatorCol", validationIndicatorCol)
  if (!is.null(initScoreCol)) stage$set("initScoreCol", initScoreCol)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(numIterations)) stage$set("numIterations", numIterations)
  if (!is.null(learningRate)) stage$set("learningRate", learningRate)
  if (!is.null(numLeaves)) stage$set("numLeaves", numLeaves)
  if (!is.null(maxDepth)) stage$set("maxDepth", maxDepth)
  if (!is.null(maxBin)) stage$set("maxBin", maxBin)
  if (!is.null(lambdaL1)) stage$set("lambdaL1", lambdaL1)
  if (!is.null(lambdaL2)) stage$set("lambdaL2", lambdaL2)
  if (!is.null(minDataInLeaf)) stage$set("minDataInLeaf", minDataInLeaf)
  if (!is.null(minSumHessianInLeaf)) stage$set("minSumHessianInLeaf", minSumHessianInLeaf)
  if (!is.null(minGainToSplit)) stage$set("minGainToSplit", minGainToSplit)
  if (!is.null(featureFraction)) stage$set("featureFraction", featureFraction)
  if (!is.null(baggingFraction)) stage$set("baggingFraction", baggingFraction)
  if (!is.null(baggingFreq)) stage$set("baggingFreq", baggingFreq)
  if (!is.null(baggingSeed)) stage$set("baggingSeed", baggingSeed)
  if (!is.null(boostingType)) stage$set("boostingType", boostingType)
  if (!is.null(topRate)) stage$set("topRate", topRate)
  if (!is.null(otherRate)) stage$set("otherRate", otherRate)
  if (!is.null(dropRate)) stage$set("dropRate", dropRate)
  if (!is.null(skipDrop)) stage$set("skipDrop", skipDrop)
  if (!is.null(maxDrop)) stage$set("maxDrop", maxDrop)
  if (!is.null(maxDeltaStep)) stage$set("maxDeltaStep", maxDeltaStep)
  if (!is.null(earlyStoppingRound)) stage$set("earlyStoppingRound", earlyStoppingRound)
  if (!is.null(objective)) stage$set("objective", objective)
  if (!is.null(metric)) stage$set("metric", metric)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(numBatches)) stage$set("numBatches", numBatches)
  if (!is.null(verbosity)) stage$set("verbosity", verbosity)
  if (!is.null(isProvideTrainingMetric)) stage$set("isProvideTrainingMetric", isProvideTrainingMetric)
  if (!is.null(useBarrierExecutionMode)) stage$set("useBarrierExecutionMode", useBarrierExecutionMode)
  if (!is.null(parallelism)) stage$set("parallelism", parallelism)
  if (!is.null(topK)) stage$set("topK", topK)
  if (!is.null(categoricalSlotIndexes)) stage$set("categoricalSlotIndexes", categoricalSlotIndexes)
  if (!is.null(categoricalSlotNames)) stage$set("categoricalSlotNames", categoricalSlotNames)
  if (!is.null(slotNames)) stage$set("slotNames", slotNames)
  if (!is.null(modelString)) stage$set("modelString", modelString)
  if (!is.null(lightGBMBooster)) stage$set("lightGBMBooster", lightGBMBooster)
  if (!is.null(fobj)) stage$set("fobj", fobj)
  if (!is.null(boostFromAverage)) stage$set("boostFromAverage", boostFromAverage)
  if (!is.null(improvementTolerance)) stage$set("improvementTolerance", improvementTolerance)
  if (!is.null(posBaggingFraction)) stage$set("posBaggingFraction", posBaggingFraction)
  if (!is.null(negBaggingFraction)) stage$set("negBaggingFraction", negBaggingFraction)
  if (!is.null(binSampleCount)) stage$set("binSampleCount", binSampleCount)
  if (!is.null(maxBinByFeature)) stage$set("maxBinByFeature", maxBinByFeature)
  if (!is.null(uniformDrop)) stage$set("uniformDrop", uniformDrop)
  if (!is.null(xgboostDartMode)) stage$set("xgboostDartMode", xgboostDartMode)
  if (!is.null(startIteration)) stage$set("startIteration", startIteration)
  if (!is.null(leafPredictionCol)) stage$set("leafPredictionCol", leafPredictionCol)
  if (!is.null(featuresShapCol)) stage$set("featuresShapCol", featuresShapCol)
  if (!is.null(chunkSize)) stage$set("chunkSize", chunkSize)
  if (!is.null(defaultListenPort)) stage$set("defaultListenPort", defaultListenPort)
  if (!is.null(driverListenPort)) stage$set("driverListenPort", driverListenPort)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(numTasks)) stage$set("numTasks", numTasks)
  if (!is.null(numThreads)) stage$set("numThreads", numThreads)
  if (!is.null(useSingleDatasetMode)) stage$set("useSingleDatasetMode", useSingleDatasetMode)
  if (!is.null(matrixType)) stage$set("matrixType", matrixType)
  if (!is.null(checkpointDir)) stage$set("checkpointDir", checkpointDir)
  if (!is.null(checkpointInterval)) stage$set("checkpointInterval", checkpointInterval)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(alpha)) stage$set("alpha", alpha)
  if (!is.null(tweedieVariancePower)) stage$set("tweedieVariancePower", tweedieVariancePower)
  stage
}

ml_linear_scalar_scaler <- function(inputCol = NULL, partitionKey = NULL, outputCol = NULL, minRequiredValue = NULL, maxRequiredValue = NULL) {
  stage <- mmlspark_amd$models$cyber$LinearScalarScaler()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(partitionKey)) stage$set("partitionKey", partitionKey)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(minRequiredValue)) stage$set("minRequiredValue", minRequiredValue)
  if (!is.null(maxRequiredValue)) stage$set("maxRequiredValue", maxRequiredValue)
  stage
}

ml_linear_scalar_scaler_model <- function(inputCol = NULL, partitionKey = NULL, outputCol = NULL, minRequiredValue = NULL, maxRequiredValue = NULL, stats = NULL) {
  stage <- mmlspark_amd$models$cyber$LinearScalarScalerModel()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(partitionKey)) stage$set("partitionKey", partitionKey)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(minRequiredValue)) stage$set("minRequiredValue", minRequiredValue)
  if (!is.null(maxRequiredValue)) stage$set("maxRequiredValue", maxRequiredValue)
  if (!is.null(stats)) stage$set("stats", stats)
  stage
}

ml_list_custom_models <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$ListCustomModels()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  stage
}

ml_multi_column_adapter <- function(baseStage = NULL, inputCols = NULL, outputCols = NULL) {
  stage <- mmlspark_amd$stages$basic$MultiColumnAdapter()
  if (!is.null(baseStage)) stage$set("baseStage", baseStage)
  if (!is.null(inputCols)) stage$set("inputCols", inputCols)
  if (!is.null(outputCols)) stage$set("outputCols", outputCols)
  stage
}

ml_multi_n_gram <- function(inputCol = NULL, outputCol = NULL, lengths = NULL) {
  stage <- mmlspark_amd$stages$text$MultiNGram()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(lengths)) stage$set("lengths", lengths)
  stage
}

ml_ner <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, language = NULL, languageCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$NER()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(language)) stage$set("language", language)
  if (!is.null(languageCol)) stage$set("languageCol", languageCol)
  stage
}

ml_nerv2 <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, language = NULL, languageCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$NERV2()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(language)) stage$set("language", language)
  if (!is.null(languageCol)) stage$set("languageCol", languageCol)
  stage
}

ml_ocr <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$OCR()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  stage
}

ml_page_splitter <- function(inputCol = NULL, outputCol = NULL, maximumPageLength = NULL, minimumPageLength = NULL, boundaryRegex = NULL) {
  stage <- mmlspark_amd$stages$text$PageSplitter()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(maximumPageLength)) stage$set("maximumPageLength", maximumPageLength)
  if (!is.null(minimumPageLength)) stage$set("minimumPageLength", minimumPageLength)
  if (!is.null(boundaryRegex)) stage$set("boundaryRegex", boundaryRegex)
  stage
}

ml_partition_consolidator <- function(concurrency = NULL, timeout = NULL) {
  stage <- mmlspark_amd$stages$batching$PartitionConsolidator()
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  stage
}

ml_pipeline <- function(stages = NULL) {
  stage <- mmlspark_amd$core$pipeline$Pipeline()
  if (!is.null(stages)) stage$set("stages", stages)
  stage
}

ml_pipeline_model <- function(stages = NULL) {
  stage <- mmlspark_amd$core$pipeline$PipelineModel()
  if (!is.null(stages)) stage$set("stages", stages)
  stage
}

ml_power_bi_writer <- function(url = NULL, batchSize = NULL, concurrency = NULL) {
  stage <- mmlspark_amd$io_http$files$PowerBIWriter()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  stage
}

ml_ranking_adapter <- function(recommenderModel = NULL, userCol = NULL, itemCol = NULL, k = NULL) {
  stage <- mmlspark_amd$models$sar$RankingAdapter()
  if (!is.null(recommenderModel)) stage$set("recommenderModel", recommenderModel)
  if (!is.null(userCol)) stage$set("userCol", userCol)
  if (!is.null(itemCol)) stage$set("itemCol", itemCol)
  if (!is.null(k)) stage$set("k", k)
  stage
}

ml_ranking_evaluator <- function(k = NULL, metricName = NULL, predictionCol = NULL, labelCol = NULL) {
  stage <- mmlspark_amd$models$sar$RankingEvaluator()
  if (!is.null(k)) stage$set("k", k)
  if (!is.null(metricName)) stage$set("metricName", metricName)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  stage
}

ml_ranking_train_validation_split <- function(estimator = NULL, userCol = NULL, itemCol = NULL, ratingCol = NULL, trainRatio = NULL, minRatingsPerUser = NULL, minRatingsPerItem = NULL, k = NULL, seed = NULL) {
  stage <- mmlspark_amd$models$sar$RankingTrainValidationSplit()
  if (!is.null(estimator)) stage$set("estimator", estimator)
  if (!is.null(userCol)) stage$set("userCol", userCol)
  if (!is.null(itemCol)) stage$set("itemCol", itemCol)
  if (!is.null(ratingCol)) stage$set("ratingCol", ratingCol)
  if (!is.null(trainRatio)) stage$set("trainRatio", trainRatio)
  if (!is.null(minRatingsPerUser)) stage$set("minRatingsPerUser", minRatingsPerUser)
  if (!is.null(minRatingsPerItem)) stage$set("minRatingsPerItem", minRatingsPerItem)
  if (!is.null(k)) stage$set("k", k)
  if (!is.null(seed)) stage$set("seed", seed)
  stage
}

ml_ranking_train_validation_split_model <- function(bestModel = NULL, validationMetric = NULL) {
  stage <- mmlspark_amd$models$sar$RankingTrainValidationSplitModel()
  if (!is.null(bestModel)) stage$set("bestModel", bestModel)
  if (!is.null(validationMetric)) stage$set("validationMetric", validationMetric)
  stage
}

ml_read_image <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$ReadImage()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  stage
}

ml_recognize_domain_specific_content <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL, model = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$RecognizeDomainSpecificContent()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  if (!is.null(model)) stage$set("model", model)
  stage
}

ml_recognize_text <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$RecognizeText()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  stage
}

ml_recommendation_indexer <- function(userInputCol = NULL, itemInputCol = NULL, userOutputCol = NULL, itemOutputCol = NULL) {
  stage <- mmlspark_amd$models$sar$RecommendationIndexer()
  if (!is.null(userInputCol)) stage$set("userInputCol", userInputCol)
  if (!is.null(itemInputCol)) stage$set("itemInputCol", itemInputCol)
  if (!is.null(userOutputCol)) stage$set("userOutputCol", userOutputCol)
  if (!is.null(itemOutputCol)) stage$set("itemOutputCol", itemOutputCol)
  stage
}

ml_recommendation_indexer_model <- function(userInputCol = NULL, itemInputCol = NULL, userOutputCol = NULL, itemOutputCol = NULL, userMap = NULL, itemMap = NULL) {
  stage <- mmlspark_amd$models$sar$RecommendationIndexerModel()
  if (!is.null(userInputCol)) stage$set("userInputCol", userInputCol)
  if (!is.null(itemInputCol)) stage$set("itemInputCol", itemInputCol)
  if (!is.null(userOutputCol)) stage$set("userOutputCol", userOutputCol)
  if (!is.null(itemOutputCol)) stage$set("itemOutputCol", itemOutputCol)
  if (!is.null(userMap)) stage$set("userMap", userMap)
  if (!is.null(itemMap)) stage$set("itemMap", itemMap)
  stage
}

ml_rename_column <- function(inputCol = NULL, outputCol = NULL) {
  stage <- mmlspark_amd$stages$basic$RenameColumn()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  stage
}

ml_repartition <- function(n = NULL, disable = NULL) {
  stage <- mmlspark_amd$stages$basic$Repartition()
  if (!is.null(n)) stage$set("n", n)
  if (!is.null(disable)) stage$set("disable", disable)
  stage
}

ml_resize_image_transformer <- function(inputCol = NULL, outputCol = NULL, height = NULL, width = NULL) {
  stage <- mmlspark_amd$models$images$ResizeImageTransformer()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(height)) stage$set("height", height)
  if (!is.null(width)) stage$set("width", width)
  stage
}

ml_sar <- function(userCol = NULL, itemCol = NULL, ratingCol = NULL, timeCol = NULL, supportThreshold = NULL, similarityFunction = NULL, timeDecayCoeff = NULL, startTime = NULL, startTimeFormat = NULL, activityTimeFormat = NULL) {
  stage <- mmlspark_amd$models$sar$SAR()
  if (!is.null(userCol)) stage$set("userCol", userCol)
  if (!is.null(itemCol)) stage$set("itemCol", itemCol)
  if (!is.null(ratingCol)) stage$set("ratingCol", ratingCol)
  if (!is.null(timeCol)) stage$set("timeCol", timeCol)
  if (!is.null(supportThreshold)) stage$set("supportThreshold", supportThreshold)
  if (!is.null(similarityFunction)) stage$set("similarityFunction", similarityFunction)
  if (!is.null(timeDecayCoeff)) stage$set("timeDecayCoeff", timeDecayCoeff)
  if (!is.null(startTime)) stage$set("startTime", startTime)
  if (!is.null(startTimeFormat)) stage$set("startTimeFormat", startTimeFormat)
  if (!is.null(activityTimeFormat)) stage$set("activityTimeFormat", activityTimeFormat)
  stage
}

ml_sar_model <- function(userCol = NULL, itemCol = NULL, ratingCol = NULL, predictionCol = NULL, sarArrays = NULL) {
  stage <- mmlspark_amd$models$sar$SARModel()
  if (!is.null(userCol)) stage$set("userCol", userCol)
  if (!is.null(itemCol)) stage$set("itemCol", itemCol)
  if (!is.null(ratingCol)) stage$set("ratingCol", ratingCol)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(sarArrays)) stage$set("sarArrays", sarArrays)
  stage
}

ml_select_columns <- function(cols = NULL) {
  stage <- mmlspark_amd$stages$basic$SelectColumns()
  if (!is.null(cols)) stage$set("cols", cols)
  stage
}

ml_simple_detect_anomalies <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, seriesCol = NULL, granularity = NULL, sensitivity = NULL, groupbyCol = NULL, timestampCol = NULL, valueCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$SimpleDetectAnomalies()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(seriesCol)) stage$set("seriesCol", seriesCol)
  if (!is.null(granularity)) stage$set("granularity", granularity)
  if (!is.null(sensitivity)) stage$set("sensitivity", sensitivity)
  if (!is.null(groupbyCol)) stage$set("groupbyCol", groupbyCol)
  if (!is.null(timestampCol)) stage$set("timestampCol", timestampCol)
  if (!is.null(valueCol)) stage$set("valueCol", valueCol)
  stage
}

ml_simple_http_transformer <- function(inputCol = NULL, outputCol = NULL, url = NULL, concurrency = NULL, flattenOutputBatches = NULL, errorCol = NULL) {
  stage <- mmlspark_amd$io_http$client$SimpleHTTPTransformer()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(flattenOutputBatches)) stage$set("flattenOutputBatches", flattenOutputBatches)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  stage
}

ml_speech_to_text <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, audioDataCol = NULL, language = NULL, format = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$SpeechToText()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(audioDataCol)) stage$set("audioDataCol", audioDataCol)
  if (!is.null(language)) stage$set("language", language)
  if (!is.null(format)) stage$set("format", format)
  stage
}

ml_speech_to_text_sdk <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, audioBytesCol = NULL, format = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$SpeechToTextSDK()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(audioBytesCol)) stage$set("audioBytesCol", audioBytesCol)
  if (!is.null(format)) stage$set("format", format)
  stage
}

ml_standard_scalar_scaler <- function(inputCol = NULL, partitionKey = NULL, outputCol = NULL, coefficientFactor = NULL) {
  stage <- mmlspark_amd$models$cyber$StandardScalarScaler()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(partitionKey)) stage$set("partitionKey", partitionKey)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(coefficientFactor)) stage$set("coefficientFactor", coefficientFactor)
  stage
}

ml_standard_scalar_scaler_model <- function(inputCol = NULL, partitionKey = NULL, outputCol = NULL, coefficientFactor = NULL, stats = NULL) {
  stage <- mmlspark_amd$models$cyber$StandardScalarScalerModel()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(partitionKey)) stage$set("partitionKey", partitionKey)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(coefficientFactor)) stage$set("coefficientFactor", coefficientFactor)
  if (!is.null(stats)) stage$set("stats", stats)
  stage
}

ml_stratified_repartition <- function(labelCol = NULL, mode = NULL, seed = NULL) {
  stage <- mmlspark_amd$stages$basic$StratifiedRepartition()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(mode)) stage$set("mode", mode)
  if (!is.null(seed)) stage$set("seed", seed)
  stage
}

ml_string_output_parser <- function(inputCol = NULL, outputCol = NULL) {
  stage <- mmlspark_amd$io_http$client$StringOutputParser()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  stage
}

ml_summarize_data <- function(counts = NULL, basic = NULL, percentiles = NULL) {
  stage <- mmlspark_amd$stages$basic$SummarizeData()
  if (!is.null(counts)) stage$set("counts", counts)
  if (!is.null(basic)) stage$set("basic", basic)
  if (!is.null(percentiles)) stage$set("percentiles", percentiles)
  stage
}

ml_superpixel_transformer <- function(model = NULL, targetCol = NULL, targetClasses = NULL, outputCol = NULL, numSamples = NULL, metricsCol = NULL, seed = NULL, rowBatch = NULL, kernelWidth = NULL, regularization = NULL, inputCol = NULL, cellSize = NULL, modifier = NULL) {
  stage <- mmlspark_amd$explainers$lime$SuperpixelTransformer()
  if (!is.null(model)) stage$set("model", model)
  if (!is.null(targetCol)) stage$set("targetCol", targetCol)
  if (!is.null(targetClasses)) stage$set("targetClasses", targetClasses)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(numSamples)) stage$set("numSamples", numSamples)
  if (!is.null(metricsCol)) stage$set("metricsCol", metricsCol)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(rowBatch)) stage$set("rowBatch", rowBatch)
  if (!is.null(kernelWidth)) stage$set("kernelWidth", kernelWidth)
  if (!is.null(regularization)) stage$set("regularization", regularization)
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(cellSize)) stage$set("cellSize", cellSize)
  if (!is.null(modifier)) stage$set("modifier", modifier)
  stage
}

ml_tabular_lime <- function(model = NULL, targetCol = NULL, targetClasses = NULL, outputCol = NULL, numSamples = NULL, metricsCol = NULL, seed = NULL, rowBatch = NULL, kernelWidth = NULL, regularization = NULL, inputCols = NULL, backgroundData = NULL) {
  stage <- mmlspark_amd$explainers$lime$TabularLIME()
  if (!is.null(model)) stage$set("model", model)
  if (!is.null(targetCol)) stage$set("targetCol", targetCol)
  if (!is.null(targetClasses)) stage$set("targetClasses", targetClasses)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(numSamples)) stage$set("numSamples", numSamples)
  if (!is.null(metricsCol)) stage$set("metricsCol", metricsCol)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(rowBatch)) stage$set("rowBatch", rowBatch)
  if (!is.null(kernelWidth)) stage$set("kernelWidth", kernelWidth)
  if (!is.null(regularization)) stage$set("regularization", regularization)
  if (!is.null(inputCols)) stage$set("inputCols", inputCols)
  if (!is.null(backgroundData)) stage$set("backgroundData", backgroundData)
  stage
}

ml_tabular_shap <- function(model = NULL, targetCol = NULL, targetClasses = NULL, outputCol = NULL, numSamples = NULL, metricsCol = NULL, seed = NULL, rowBatch = NULL, inputCols = NULL, backgroundData = NULL) {
  stage <- mmlspark_amd$explainers$shap$TabularSHAP()
  if (!is.null(model)) stage$set("model", model)
  if (!is.null(targetCol)) stage$set("targetCol", targetCol)
  if (!is.null(targetClasses)) stage$set("targetClasses", targetClasses)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(numSamples)) stage$set("numSamples", numSamples)
  if (!is.null(metricsCol)) stage$set("metricsCol", metricsCol)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(rowBatch)) stage$set("rowBatch", rowBatch)
  if (!is.null(inputCols)) stage$set("inputCols", inputCols)
  if (!is.null(backgroundData)) stage$set("backgroundData", backgroundData)
  stage
}

ml_tag_image <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, imageUrlCol = NULL, imageBytesCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$TagImage()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(imageUrlCol)) stage$set("imageUrlCol", imageUrlCol)
  if (!is.null(imageBytesCol)) stage$set("imageBytesCol", imageBytesCol)
  stage
}

ml_text_featurizer <- function(inputCol = NULL, outputCol = NULL, useTokenizer = NULL, tokenizerPattern = NULL, toLowercase = NULL, minTokenLength = NULL, useNGram = NULL, nGramLength = NULL, numFeatures = NULL, useIDF = NULL, minDocFreq = NULL, useStopWordsRemover = NULL, stopWords = NULL, caseSensitiveStopWords = NULL, binary = NULL) {
  stage <- mmlspark_amd$stages$text$TextFeaturizer()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(useTokenizer)) stage$set("useTokenizer", useTokenizer)
  if (!is.null(tokenizerPattern)) stage$set("tokenizerPattern", tokenizerPattern)
  if (!is.null(toLowercase)) stage$set("toLowercase", toLowercase)
  if (!is.null(minTokenLength)) stage$set("minTokenLength", minTokenLength)
  if (!is.null(useNGram)) stage$set("useNGram", useNGram)
  if (!is.null(nGramLength)) stage$set("nGramLength", nGramLength)
  if (!is.null(numFeatures)) stage$set("numFeatures", numFeatures)
  if (!is.null(useIDF)) stage$set("useIDF", useIDF)
  if (!is.null(minDocFreq)) stage$set("minDocFreq", minDocFreq)
  if (!is.null(useStopWordsRemover)) stage$set("useStopWordsRemover", useStopWordsRemover)
  if (!is.null(stopWords)) stage$set("stopWords", stopWords)
  if (!is.null(caseSensitiveStopWords)) stage$set("caseSensitiveStopWords", caseSensitiveStopWords)
  if (!is.null(binary)) stage$set("binary", binary)
  stage
}

ml_text_featurizer_model <- function(inputCol = NULL, outputCol = NULL, useTokenizer = NULL, tokenizerPattern = NULL, toLowercase = NULL, minTokenLength = NULL, useNGram = NULL, nGramLength = NULL, numFeatures = NULL, useStopWordsRemover = NULL, stopWords = NULL, caseSensitiveStopWords = NULL, binary = NULL, idf = NULL) {
  stage <- mmlspark_amd$stages$text$TextFeaturizerModel()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(useTokenizer)) stage$set("useTokenizer", useTokenizer)
  if (!is.null(tokenizerPattern)) stage$set("tokenizerPattern", tokenizerPattern)
  if (!is.null(toLowercase)) stage$set("toLowercase", toLowercase)
  if (!is.null(minTokenLength)) stage$set("minTokenLength", minTokenLength)
  if (!is.null(useNGram)) stage$set("useNGram", useNGram)
  if (!is.null(nGramLength)) stage$set("nGramLength", nGramLength)
  if (!is.null(numFeatures)) stage$set("numFeatures", numFeatures)
  if (!is.null(useStopWordsRemover)) stage$set("useStopWordsRemover", useStopWordsRemover)
  if (!is.null(stopWords)) stage$set("stopWords", stopWords)
  if (!is.null(caseSensitiveStopWords)) stage$set("caseSensitiveStopWords", caseSensitiveStopWords)
  if (!is.null(binary)) stage$set("binary", binary)
  if (!is.null(idf)) stage$set("idf", idf)
  stage
}

ml_text_lime <- function(model = NULL, targetCol = NULL, targetClasses = NULL, outputCol = NULL, numSamples = NULL, metricsCol = NULL, seed = NULL, rowBatch = NULL, kernelWidth = NULL, regularization = NULL, inputCol = NULL, tokensCol = NULL) {
  stage <- mmlspark_amd$explainers$lime$TextLIME()
  if (!is.null(model)) stage$set("model", model)
  if (!is.null(targetCol)) stage$set("targetCol", targetCol)
  if (!is.null(targetClasses)) stage$set("targetClasses", targetClasses)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(numSamples)) stage$set("numSamples", numSamples)
  if (!is.null(metricsCol)) stage$set("metricsCol", metricsCol)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(rowBatch)) stage$set("rowBatch", rowBatch)
  if (!is.null(kernelWidth)) stage$set("kernelWidth", kernelWidth)
  if (!is.null(regularization)) stage$set("regularization", regularization)
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(tokensCol)) stage$set("tokensCol", tokensCol)
  stage
}

ml_text_preprocessor <- function(inputCol = NULL, outputCol = NULL, normFunc = NULL, map = NULL) {
  stage <- mmlspark_amd$stages$basic$TextPreprocessor()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(normFunc)) stage$set("normFunc", normFunc)
  if (!is.null(map)) stage$set("map", map)
  stage
}

ml_text_shap <- function(model = NULL, targetCol = NULL, targetClasses = NULL, outputCol = NULL, numSamples = NULL, metricsCol = NULL, seed = NULL, rowBatch = NULL, inputCol = NULL, tokensCol = NULL) {
  stage <- mmlspark_amd$explainers$shap$TextSHAP()
  if (!is.null(model)) stage$set("model", model)
  if (!is.null(targetCol)) stage$set("targetCol", targetCol)
  if (!is.null(targetClasses)) stage$set("targetClasses", targetClasses)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(numSamples)) stage$set("numSamples", numSamples)
  if (!is.null(metricsCol)) stage$set("metricsCol", metricsCol)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(rowBatch)) stage$set("rowBatch", rowBatch)
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(tokensCol)) stage$set("tokensCol", tokensCol)
  stage
}

ml_text_sentiment <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, language = NULL, languageCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$TextSentiment()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(language)) stage$set("language", language)
  if (!is.null(languageCol)) stage$set("languageCol", languageCol)
  stage
}

ml_text_sentiment_v2 <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, language = NULL, languageCol = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$TextSentimentV2()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(language)) stage$set("language", language)
  if (!is.null(languageCol)) stage$set("languageCol", languageCol)
  stage
}

ml_time_interval_mini_batch_transformer <- function(millisToWait = NULL, maxBatchSize = NULL) {
  stage <- mmlspark_amd$stages$batching$TimeIntervalMiniBatchTransformer()
  if (!is.null(millisToWait)) stage$set("millisToWait", millisToWait)
  if (!is.null(maxBatchSize)) stage$set("maxBatchSize", maxBatchSize)
  stage
}

ml_timer <- function(stage = NULL, logToScala = NULL, disableMaterialization = NULL) {
  stage <- mmlspark_amd$stages$basic$Timer()
  if (!is.null(stage)) stage$set("stage", stage)
  if (!is.null(logToScala)) stage$set("logToScala", logToScala)
  if (!is.null(disableMaterialization)) stage$set("disableMaterialization", disableMaterialization)
  stage
}

ml_torch_model <- function(inputCol = NULL, outputCol = NULL, batchSize = NULL, moduleBytes = NULL, device = NULL) {
  stage <- mmlspark_amd$models$image_featurizer$TorchModel()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(moduleBytes)) stage$set("moduleBytes", moduleBytes)
  if (!is.null(device)) stage$set("device", device)
  stage
}

ml_train_classifier <- function(model = NULL, labelCol = NULL, featuresCol = NULL, numFeatures = NULL, reindexLabel = NULL) {
  stage <- mmlspark_amd$stages$train$TrainClassifier()
  if (!is.null(model)) stage$set("model", model)
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(numFeatures)) stage$set("numFeatures", numFeatures)
  if (!is.null(reindexLabel)) stage$set("reindexLabel", reindexLabel)
  stage
}

ml_train_regressor <- function(model = NULL, labelCol = NULL, featuresCol = NULL) {
  stage <- mmlspark_amd$stages$train$TrainRegressor()
  if (!is.null(model)) stage$set("model", model)
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  stage
}

ml_trained_classifier_model <- function(labelCol = NULL, levels = NULL, featuresCol = NULL, featurizerModel = NULL, innerModel = NULL) {
  stage <- mmlspark_amd$stages$train$TrainedClassifierModel()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(levels)) stage$set("levels", levels)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(featurizerModel)) stage$set("featurizerModel", featurizerModel)
  if (!is.null(innerModel)) stage$set("innerModel", innerModel)
  stage
}

ml_trained_regressor_model <- function(labelCol = NULL, featurizerModel = NULL, innerModel = NULL) {
  stage <- mmlspark_amd$stages$train$TrainedRegressorModel()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featurizerModel)) stage$set("featurizerModel", featurizerModel)
  if (!is.null(innerModel)) stage$set("innerModel", innerModel)
  stage
}

ml_translate <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, toLanguage = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$Translate()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(toLanguage)) stage$set("toLanguage", toLanguage)
  stage
}

ml_transliterate <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, textCol = NULL, toLanguage = NULL, fromScript = NULL, toScript = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$Transliterate()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(textCol)) stage$set("textCol", textCol)
  if (!is.null(toLanguage)) stage$set("toLanguage", toLanguage)
  if (!is.null(fromScript)) stage$set("fromScript", fromScript)
  if (!is.null(toScript)) stage$set("toScript", toScript)
  stage
}

ml_tune_hyperparameters <- function(evaluationMetric = NULL, numFolds = NULL, numRuns = NULL, searchMode = NULL, parallelism = NULL, seed = NULL, labelCol = NULL, models = NULL, paramSpace = NULL) {
  stage <- mmlspark_amd$stages$automl$TuneHyperparameters()
  if (!is.null(evaluationMetric)) stage$set("evaluationMetric", evaluationMetric)
  if (!is.null(numFolds)) stage$set("numFolds", numFolds)
  if (!is.null(numRuns)) stage$set("numRuns", numRuns)
  if (!is.null(searchMode)) stage$set("searchMode", searchMode)
  if (!is.null(parallelism)) stage$set("parallelism", parallelism)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(models)) stage$set("models", models)
  if (!is.null(paramSpace)) stage$set("paramSpace", paramSpace)
  stage
}

ml_tune_hyperparameters_model <- function(bestModel = NULL, bestMetric = NULL, bestParams = NULL) {
  stage <- mmlspark_amd$stages$automl$TuneHyperparametersModel()
  if (!is.null(bestModel)) stage$set("bestModel", bestModel)
  if (!is.null(bestMetric)) stage$set("bestMetric", bestMetric)
  if (!is.null(bestParams)) stage$set("bestParams", bestParams)
  stage
}

ml_udf_transformer <- function(inputCol = NULL, inputCols = NULL, outputCol = NULL) {
  stage <- mmlspark_amd$stages$basic$UDFTransformer()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(inputCols)) stage$set("inputCols", inputCols)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  stage
}

ml_unicode_normalize <- function(inputCol = NULL, outputCol = NULL, form = NULL, lower = NULL) {
  stage <- mmlspark_amd$stages$basic$UnicodeNormalize()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(form)) stage$set("form", form)
  if (!is.null(lower)) stage$set("lower", lower)
  stage
}

ml_unroll_binary_image <- function(inputCol = NULL, outputCol = NULL, height = NULL, width = NULL) {
  stage <- mmlspark_amd$models$images$UnrollBinaryImage()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(height)) stage$set("height", height)
  if (!is.null(width)) stage$set("width", width)
  stage
}

ml_unroll_image <- function(inputCol = NULL, outputCol = NULL) {
  stage <- mmlspark_amd$models$images$UnrollImage()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  stage
}

ml_value_indexer <- function(inputCol = NULL, outputCol = NULL) {
  stage <- mmlspark_amd$stages$featurize$ValueIndexer()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  stage
}

ml_value_indexer_model <- function(inputCol = NULL, outputCol = NULL, levels = NULL) {
  stage <- mmlspark_amd$stages$featurize$ValueIndexerModel()
  if (!is.null(inputCol)) stage$set("inputCol", inputCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(levels)) stage$set("levels", levels)
  stage
}

ml_vector_lime <- function(model = NULL, targetCol = NULL, targetClasses = NULL, outputCol = NULL, numSamples = NULL, metricsCol = NULL, seed = NULL, rowBatch = NULL, kernelWidth = NULL, regularization = NULL, featuresCol = NULL, backgroundData = NULL) {
  stage <- mmlspark_amd$explainers$lime$VectorLIME()
  if (!is.null(model)) stage$set("model", model)
  if (!is.null(targetCol)) stage$set("targetCol", targetCol)
  if (!is.null(targetClasses)) stage$set("targetClasses", targetClasses)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(numSamples)) stage$set("numSamples", numSamples)
  if (!is.null(metricsCol)) stage$set("metricsCol", metricsCol)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(rowBatch)) stage$set("rowBatch", rowBatch)
  if (!is.null(kernelWidth)) stage$set("kernelWidth", kernelWidth)
  if (!is.null(regularization)) stage$set("regularization", regularization)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(backgroundData)) stage$set("backgroundData", backgroundData)
  stage
}

ml_vector_shap <- function(model = NULL, targetCol = NULL, targetClasses = NULL, outputCol = NULL, numSamples = NULL, metricsCol = NULL, seed = NULL, rowBatch = NULL, featuresCol = NULL, backgroundData = NULL) {
  stage <- mmlspark_amd$explainers$shap$VectorSHAP()
  if (!is.null(model)) stage$set("model", model)
  if (!is.null(targetCol)) stage$set("targetCol", targetCol)
  if (!is.null(targetClasses)) stage$set("targetClasses", targetClasses)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(numSamples)) stage$set("numSamples", numSamples)
  if (!is.null(metricsCol)) stage$set("metricsCol", metricsCol)
  if (!is.null(seed)) stage$set("seed", seed)
  if (!is.null(rowBatch)) stage$set("rowBatch", rowBatch)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(backgroundData)) stage$set("backgroundData", backgroundData)
  stage
}

ml_verify_faces <- function(url = NULL, subscriptionKey = NULL, subscriptionKeyCol = NULL, outputCol = NULL, errorCol = NULL, concurrency = NULL, timeout = NULL, faceId1Col = NULL, faceId2Col = NULL) {
  stage <- mmlspark_amd$io_http$cognitive$VerifyFaces()
  if (!is.null(url)) stage$set("url", url)
  if (!is.null(subscriptionKey)) stage$set("subscriptionKey", subscriptionKey)
  if (!is.null(subscriptionKeyCol)) stage$set("subscriptionKeyCol", subscriptionKeyCol)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(errorCol)) stage$set("errorCol", errorCol)
  if (!is.null(concurrency)) stage$set("concurrency", concurrency)
  if (!is.null(timeout)) stage$set("timeout", timeout)
  if (!is.null(faceId1Col)) stage$set("faceId1Col", faceId1Col)
  if (!is.null(faceId2Col)) stage$set("faceId2Col", faceId2Col)
  stage
}

ml_vowpal_wabbit_classification_model <- function(labelCol = NULL, featuresCol = NULL, additionalFeatures = NULL, weightCol = NULL, predictionCol = NULL, numPasses = NULL, learningRate = NULL, powerT = NULL, l1 = NULL, l2 = NULL, numBits = NULL, lossFunction = NULL, batchSize = NULL, hashSeed = NULL, adaptive = NULL, normalized = NULL, invariant = NULL, initialModel = NULL, passThroughArgs = NULL, args = NULL, testArgs = NULL, interactions = NULL, ignoreNamespaces = NULL, useBarrierExecutionMode = NULL, bfgs = NULL, holdoutOff = NULL, holdoutPeriod = NULL, earlyTerminate = NULL, maxIterBfgs = NULL, device = NULL, weightsArrays = NULL, rawPredictionCol = NULL, probabilityCol = NULL, labelConversion = NULL) {
  stage <- mmlspark_amd$models$vw$estimators$VowpalWabbitClassificationModel()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(additionalFeatures)) stage$set("additionalFeatures", additionalFeatures)
  if (!is.null(weightCol)) stage$set("weightCol", weightCol)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(numPasses)) stage$set("numPasses", numPasses)
  if (!is.null(learningRate)) stage$set("learningRate", learningRate)
  if (!is.null(powerT)) stage$set("powerT", powerT)
  if (!is.null(l1)) stage$set("l1", l1)
  if (!is.null(l2)) stage$set("l2", l2)
  if (!is.null(numBits)) stage$set("numBits", numBits)
  if (!is.null(lossFunction)) stage$set("lossFunction", lossFunction)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(hashSeed)) stage$set("hashSeed", hashSeed)
  if (!is.null(adaptive)) stage$set("adaptive", adaptive)
  if (!is.null(normalized)) stage$set("normalized", normalized)
  if (!is.null(invariant)) stage$set("invariant", invariant)
  if (!is.null(initialModel)) stage$set("initialModel", initialModel)
  if (!is.null(passThroughArgs)) stage$set("passThroughArgs", passThroughArgs)
  if (!is.null(args)) stage$set("args", args)
  if (!is.null(testArgs)) stage$set("testArgs", testArgs)
  if (!is.null(interactions)) stage$set("interactions", interactions)
  if (!is.null(ignoreNamespaces)) stage$set("ignoreNamespaces", ignoreNamespaces)
  if (!is.null(useBarrierExecutionMode)) stage$set("useBarrierExecutionMode", useBarrierExecutionMode)
  if (!is.null(bfgs)) stage$set("bfgs", bfgs)
  if (!is.null(holdoutOff)) stage$set("holdoutOff", holdoutOff)
  if (!is.null(holdoutPeriod)) stage$set("holdoutPeriod", holdoutPeriod)
  if (!is.null(earlyTerminate)) stage$set("earlyTerminate", earlyTerminate)
  if (!is.null(maxIterBfgs)) stage$set("maxIterBfgs", maxIterBfgs)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(weightsArrays)) stage$set("weightsArrays", weightsArrays)
  if (!is.null(rawPredictionCol)) stage$set("rawPredictionCol", rawPredictionCol)
  if (!is.null(probabilityCol)) stage$set("probabilityCol", probabilityCol)
  if (!is.null(labelConversion)) stage$set("labelConversion", labelConversion)
  stage
}

ml_vowpal_wabbit_classifier <- function(labelCol = NULL, featuresCol = NULL, additionalFeatures = NULL, weightCol = NULL, predictionCol = NULL, numPasses = NULL, learningRate = NULL, powerT = NULL, l1 = NULL, l2 = NULL, numBits = NULL, lossFunction = NULL, batchSize = NULL, hashSeed = NULL, adaptive = NULL, normalized = NULL, invariant = NULL, initialModel = NULL, passThroughArgs = NULL, args = NULL, testArgs = NULL, interactions = NULL, ignoreNamespaces = NULL, useBarrierExecutionMode = NULL, bfgs = NULL, holdoutOff = NULL, holdoutPeriod = NULL, earlyTerminate = NULL, maxIterBfgs = NULL, device = NULL, rawPredictionCol = NULL, probabilityCol = NULL, labelConversion = NULL) {
  stage <- mmlspark_amd$models$vw$estimators$VowpalWabbitClassifier()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(additionalFeatures)) stage$set("additionalFeatures", additionalFeatures)
  if (!is.null(weightCol)) stage$set("weightCol", weightCol)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(numPasses)) stage$set("numPasses", numPasses)
  if (!is.null(learningRate)) stage$set("learningRate", learningRate)
  if (!is.null(powerT)) stage$set("powerT", powerT)
  if (!is.null(l1)) stage$set("l1", l1)
  if (!is.null(l2)) stage$set("l2", l2)
  if (!is.null(numBits)) stage$set("numBits", numBits)
  if (!is.null(lossFunction)) stage$set("lossFunction", lossFunction)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(hashSeed)) stage$set("hashSeed", hashSeed)
  if (!is.null(adaptive)) stage$set("adaptive", adaptive)
  if (!is.null(normalized)) stage$set("normalized", normalized)
  if (!is.null(invariant)) stage$set("invariant", invariant)
  if (!is.null(initialModel)) stage$set("initialModel", initialModel)
  if (!is.null(passThroughArgs)) stage$set("passThroughArgs", passThroughArgs)
  if (!is.null(args)) stage$set("args", args)
  if (!is.null(testArgs)) stage$set("testArgs", testArgs)
  if (!is.null(interactions)) stage$set("interactions", interactions)
  if (!is.null(ignoreNamespaces)) stage$set("ignoreNamespaces", ignoreNamespaces)
  if (!is.null(useBarrierExecutionMode)) stage$set("useBarrierExecutionMode", useBarrierExecutionMode)
  if (!is.null(bfgs)) stage$set("bfgs", bfgs)
  if (!is.null(holdoutOff)) stage$set("holdoutOff", holdoutOff)
  if (!is.null(holdoutPeriod)) stage$set("holdoutPeriod", holdoutPeriod)
  if (!is.null(earlyTerminate)) stage$set("earlyTerminate", earlyTerminate)
  if (!is.null(maxIterBfgs)) stage$set("maxIterBfgs", maxIterBfgs)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(rawPredictionCol)) stage$set("rawPredictionCol", rawPredictionCol)
  if (!is.null(probabilityCol)) stage$set("probabilityCol", probabilityCol)
  if (!is.null(labelConversion)) stage$set("labelConversion", labelConversion)
  stage
}

ml_vowpal_wabbit_contextual_bandit <- function(labelCol = NULL, featuresCol = NULL, additionalFeatures = NULL, weightCol = NULL, predictionCol = NULL, numPasses = NULL, learningRate = NULL, powerT = NULL, l1 = NULL, l2 = NULL, numBits = NULL, lossFunction = NULL, batchSize = NULL, hashSeed = NULL, adaptive = NULL, normalized = NULL, invariant = NULL, initialModel = NULL, passThroughArgs = NULL, args = NULL, testArgs = NULL, interactions = NULL, ignoreNamespaces = NULL, useBarrierExecutionMode = NULL, bfgs = NULL, holdoutOff = NULL, holdoutPeriod = NULL, earlyTerminate = NULL, maxIterBfgs = NULL, device = NULL, sharedCol = NULL, additionalSharedFeatures = NULL, chosenActionCol = NULL, probabilityCol = NULL, epsilon = NULL) {
  stage <- mmlspark_amd$models$vw$estimators$VowpalWabbitContextualBandit()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(additionalFeatures)) stage$set("additionalFeatures", additionalFeatures)
  if (!is.null(weightCol)) stage$set("weightCol", weightCol)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(numPasses)) stage$set("numPasses", numPasses)
  if (!is.null(learningRate)) stage$set("learningRate", learningRate)
  if (!is.null(powerT)) stage$set("powerT", powerT)
  if (!is.null(l1)) stage$set("l1", l1)
  if (!is.null(l2)) stage$set("l2", l2)
  if (!is.null(numBits)) stage$set("numBits", numBits)
  if (!is.null(lossFunction)) stage$set("lossFunction", lossFunction)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(hashSeed)) stage$set("hashSeed", hashSeed)
  if (!is.null(adaptive)) stage$set("adaptive", adaptive)
  if (!is.null(normalized)) stage$set("normalized", normalized)
  if (!is.null(invariant)) stage$set("invariant", invariant)
  if (!is.null(initialModel)) stage$set("initialModel", initialModel)
  if (!is.null(passThroughArgs)) stage$set("passThroughArgs", passThroughArgs)
  if (!is.null(args)) stage$set("args", args)
  if (!is.null(testArgs)) stage$set("testArgs", testArgs)
  if (!is.null(interactions)) stage$set("interactions", interactions)
  if (!is.null(ignoreNamespaces)) stage$set("ignoreNamespaces", ignoreNamespaces)
  if (!is.null(useBarrierExecutionMode)) stage$set("useBarrierExecutionMode", useBarrierExecutionMode)
  if (!is.null(bfgs)) stage$set("bfgs", bfgs)
  if (!is.null(holdoutOff)) stage$set("holdoutOff", holdoutOff)
  if (!is.null(holdoutPeriod)) stage$set("holdoutPeriod", holdoutPeriod)
  if (!is.null(earlyTerminate)) stage$set("earlyTerminate", earlyTerminate)
  if (!is.null(maxIterBfgs)) stage$set("maxIterBfgs", maxIterBfgs)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(sharedCol)) stage$set("sharedCol", sharedCol)
  if (!is.null(additionalSharedFeatures)) stage$set("additionalSharedFeatures", additionalSharedFeatures)
  if (!is.null(chosenActionCol)) stage$set("chosenActionCol", chosenActionCol)
  if (!is.null(probabilityCol)) stage$set("probabilityCol", probabilityCol)
  if (!is.null(epsilon)) stage$set("epsilon", epsilon)
  stage
}

ml_vowpal_wabbit_contextual_bandit_model <- function(labelCol = NULL, featuresCol = NULL, additionalFeatures = NULL, weightCol = NULL, predictionCol = NULL, numPasses = NULL, learningRate = NULL, powerT = NULL, l1 = NULL, l2 = NULL, numBits = NULL, lossFunction = NULL, batchSize = NULL, hashSeed = NULL, adaptive = NULL, normalized = NULL, invariant = NULL, initialModel = NULL, passThroughArgs = NULL, args = NULL, testArgs = NULL, interactions = NULL, ignoreNamespaces = NULL, useBarrierExecutionMode = NULL, bfgs = NULL, holdoutOff = NULL, holdoutPeriod = NULL, earlyTerminate = NULL, maxIterBfgs = NULL, device = NULL, weightsArrays = NULL, sharedCol = NULL, additionalSharedFeatures = NULL, chosenActionCol = NULL, probabilityCol = NULL, epsilon = NULL) {
  stage <- mmlspark_amd$models$vw$estimators$VowpalWabbitContextualBanditModel()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(additionalFeatures)) stage$set("additionalFeatures", additionalFeatures)
  if (!is.null(weightCol)) stage$set("weightCol", weightCol)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(numPasses)) stage$set("numPasses", numPasses)
  if (!is.null(learningRate)) stage$set("learningRate", learningRate)
  if (!is.null(powerT)) stage$set("powerT", powerT)
  if (!is.null(l1)) stage$set("l1", l1)
  if (!is.null(l2)) stage$set("l2", l2)
  if (!is.null(numBits)) stage$set("numBits", numBits)
  if (!is.null(lossFunction)) stage$set("lossFunction", lossFunction)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(hashSeed)) stage$set("hashSeed", hashSeed)
  if (!is.null(adaptive)) stage$set("adaptive", adaptive)
  if (!is.null(normalized)) stage$set("normalized", normalized)
  if (!is.null(invariant)) stage$set("invariant", invariant)
  if (!is.null(initialModel)) stage$set("initialModel", initialModel)
  if (!is.null(passThroughArgs)) stage$set("passThroughArgs", passThroughArgs)
  if (!is.null(args)) stage$set("args", args)
  if (!is.null(testArgs)) stage$set("testArgs", testArgs)
  if (!is.null(interactions)) stage$set("interactions", interactions)
  if (!is.null(ignoreNamespaces)) stage$set("ignoreNamespaces", ignoreNamespaces)
  if (!is.null(useBarrierExecutionMode)) stage$set("useBarrierExecutionMode", useBarrierExecutionMode)
  if (!is.null(bfgs)) stage$set("bfgs", bfgs)
  if (!is.null(holdoutOff)) stage$set("holdoutOff", holdoutOff)
  if (!is.null(holdoutPeriod)) stage$set("holdoutPeriod", holdoutPeriod)
  if (!is.null(earlyTerminate)) stage$set("earlyTerminate", earlyTerminate)
  if (!is.null(maxIterBfgs)) stage$set("maxIterBfgs", maxIterBfgs)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(weightsArrays)) stage$set("weightsArrays", weightsArrays)
  if (!is.null(sharedCol)) stage$set("sharedCol", sharedCol)
  if (!is.null(additionalSharedFeatures)) stage$set("additionalSharedFeatures", additionalSharedFeatures)
  if (!is.null(chosenActionCol)) stage$set("chosenActionCol", chosenActionCol)
  if (!is.null(probabilityCol)) stage$set("probabilityCol", probabilityCol)
  if (!is.null(epsilon)) stage$set("epsilon", epsilon)
  stage
}

ml_vowpal_wabbit_featurizer <- function(inputCols = NULL, outputCol = NULL, numBits = NULL, sumCollisions = NULL, stringSplitInputCols = NULL, prefixStringsWithColumnName = NULL, preserveOrderNumBits = NULL, seed = NULL) {
  stage <- mmlspark_amd$models$vw$featurizer$VowpalWabbitFeaturizer()
  if (!is.null(inputCols)) stage$set("inputCols", inputCols)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(numBits)) stage$set("numBits", numBits)
  if (!is.null(sumCollisions)) stage$set("sumCollisions", sumCollisions)
  if (!is.null(stringSplitInputCols)) stage$set("stringSplitInputCols", stringSplitInputCols)
  if (!is.null(prefixStringsWithColumnName)) stage$set("prefixStringsWithColumnName", prefixStringsWithColumnName)
  if (!is.null(preserveOrderNumBits)) stage$set("preserveOrderNumBits", preserveOrderNumBits)
  if (!is.null(seed)) stage$set("seed", seed)
  stage
}

ml_vowpal_wabbit_interactions <- function(inputCols = NULL, outputCol = NULL, numBits = NULL, sumCollisions = NULL) {
  stage <- mmlspark_amd$models$vw$featurizer$VowpalWabbitInteractions()
  if (!is.null(inputCols)) stage$set("inputCols", inputCols)
  if (!is.null(outputCol)) stage$set("outputCol", outputCol)
  if (!is.null(numBits)) stage$set("numBits", numBits)
  if (!is.null(sumCollisions)) stage$set("sumCollisions", sumCollisions)
  stage
}

ml_vowpal_wabbit_regressor <- function(labelCol = NULL, featuresCol = NULL, additionalFeatures = NULL, weightCol = NULL, predictionCol = NULL, numPasses = NULL, learningRate = NULL, powerT = NULL, l1 = NULL, l2 = NULL, numBits = NULL, lossFunction = NULL, batchSize = NULL, hashSeed = NULL, adaptive = NULL, normalized = NULL, invariant = NULL, initialModel = NULL, passThroughArgs = NULL, args = NULL, testArgs = NULL, interactions = NULL, ignoreNamespaces = NULL, useBarrierExecutionMode = NULL, bfgs = NULL, holdoutOff = NULL, holdoutPeriod = NULL, earlyTerminate = NULL, maxIterBfgs = NULL, device = NULL) {
  stage <- mmlspark_amd$models$vw$estimators$VowpalWabbitRegressor()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(additionalFeatures)) stage$set("additionalFeatures", additionalFeatures)
  if (!is.null(weightCol)) stage$set("weightCol", weightCol)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(numPasses)) stage$set("numPasses", numPasses)
  if (!is.null(learningRate)) stage$set("learningRate", learningRate)
  if (!is.null(powerT)) stage$set("powerT", powerT)
  if (!is.null(l1)) stage$set("l1", l1)
  if (!is.null(l2)) stage$set("l2", l2)
  if (!is.null(numBits)) stage$set("numBits", numBits)
  if (!is.null(lossFunction)) stage$set("lossFunction", lossFunction)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(hashSeed)) stage$set("hashSeed", hashSeed)
  if (!is.null(adaptive)) stage$set("adaptive", adaptive)
  if (!is.null(normalized)) stage$set("normalized", normalized)
  if (!is.null(invariant)) stage$set("invariant", invariant)
  if (!is.null(initialModel)) stage$set("initialModel", initialModel)
  if (!is.null(passThroughArgs)) stage$set("passThroughArgs", passThroughArgs)
  if (!is.null(args)) stage$set("args", args)
  if (!is.null(testArgs)) stage$set("testArgs", testArgs)
  if (!is.null(interactions)) stage$set("interactions", interactions)
  if (!is.null(ignoreNamespaces)) stage$set("ignoreNamespaces", ignoreNamespaces)
  if (!is.null(useBarrierExecutionMode)) stage$set("useBarrierExecutionMode", useBarrierExecutionMode)
  if (!is.null(bfgs)) stage$set("bfgs", bfgs)
  if (!is.null(holdoutOff)) stage$set("holdoutOff", holdoutOff)
  if (!is.null(holdoutPeriod)) stage$set("holdoutPeriod", holdoutPeriod)
  if (!is.null(earlyTerminate)) stage$set("earlyTerminate", earlyTerminate)
  if (!is.null(maxIterBfgs)) stage$set("maxIterBfgs", maxIterBfgs)
  if (!is.null(device)) stage$set("device", device)
  stage
}

ml_vowpal_wabbit_regressor_model <- function(labelCol = NULL, featuresCol = NULL, additionalFeatures = NULL, weightCol = NULL, predictionCol = NULL, numPasses = NULL, learningRate = NULL, powerT = NULL, l1 = NULL, l2 = NULL, numBits = NULL, lossFunction = NULL, batchSize = NULL, hashSeed = NULL, adaptive = NULL, normalized = NULL, invariant = NULL, initialModel = NULL, passThroughArgs = NULL, args = NULL, testArgs = NULL, interactions = NULL, ignoreNamespaces = NULL, useBarrierExecutionMode = NULL, bfgs = NULL, holdoutOff = NULL, holdoutPeriod = NULL, earlyTerminate = NULL, maxIterBfgs = NULL, device = NULL, weightsArrays = NULL) {
  stage <- mmlspark_amd$models$vw$estimators$VowpalWabbitRegressorModel()
  if (!is.null(labelCol)) stage$set("labelCol", labelCol)
  if (!is.null(featuresCol)) stage$set("featuresCol", featuresCol)
  if (!is.null(additionalFeatures)) stage$set("additionalFeatures", additionalFeatures)
  if (!is.null(weightCol)) stage$set("weightCol", weightCol)
  if (!is.null(predictionCol)) stage$set("predictionCol", predictionCol)
  if (!is.null(numPasses)) stage$set("numPasses", numPasses)
  if (!is.null(learningRate)) stage$set("learningRate", learningRate)
  if (!is.null(powerT)) stage$set("powerT", powerT)
  if (!is.null(l1)) stage$set("l1", l1)
  if (!is.null(l2)) stage$set("l2", l2)
  if (!is.null(numBits)) stage$set("numBits", numBits)
  if (!is.null(lossFunction)) stage$set("lossFunction", lossFunction)
  if (!is.null(batchSize)) stage$set("batchSize", batchSize)
  if (!is.null(hashSeed)) stage$set("hashSeed", hashSeed)
  if (!is.null(adaptive)) stage$set("adaptive", adaptive)
  if (!is.null(normalized)) stage$set("normalized", normalized)
  if (!is.null(invariant)) stage$set("invariant", invariant)
  if (!is.null(initialModel)) stage$set("initialModel", initialModel)
  if (!is.null(passThroughArgs)) stage$set("passThroughArgs", passThroughArgs)
  if (!is.null(args)) stage$set("args", args)
  if (!is.null(testArgs)) stage$set("testArgs", testArgs)
  if (!is.null(interactions)) stage$set("interactions", interactions)
  if (!is.null(ignoreNamespaces)) stage$set("ignoreNamespaces", ignoreNamespaces)
  if (!is.null(useBarrierExecutionMode)) stage$set("useBarrierExecutionMode", useBarrierExecutionMode)
  if (!is.null(bfgs)) stage$set("bfgs", bfgs)
  if (!is.null(holdoutOff)) stage$set("holdoutOff", holdoutOff)
  if (!is.null(holdoutPeriod)) stage$set("holdoutPeriod", holdoutPeriod)
  if (!is.null(earlyTerminate)) stage$set("earlyTerminate", earlyTerminate)
  if (!is.null(maxIterBfgs)) stage$set("maxIterBfgs", maxIterBfgs)
  if (!is.null(device)) stage$set("device", device)
  if (!is.null(weightsArrays)) stage$set("weightsArrays", weightsArrays)
  stage
}
