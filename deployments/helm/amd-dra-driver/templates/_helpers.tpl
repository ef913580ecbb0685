{{/*
Chart name, configurable via nameOverride.
*/}}
{{- define "amd-dra-driver.name" -}}
{{- default .Chart.Name .Values.nameOverride | trunc 63 | trimSuffix "-" }}
{{- end }}

{{/*
Fully qualified name (release-scoped), configurable via fullnameOverride.
*/}}
{{- define "amd-dra-driver.fullname" -}}
{{- if .Values.fullnameOverride -}}
{{- .Values.fullnameOverride | trunc 63 | trimSuffix "-" -}}
{{- else -}}
{{- .Release.Name | trunc 63 | trimSuffix "-" -}}
{{- end -}}
{{- end }}

{{/*
Target namespace: namespaceOverride > values.namespace (legacy knob) >
release namespace.
*/}}
{{- define "amd-dra-driver.namespace" -}}
{{- default (default .Release.Namespace .Values.namespace) .Values.namespaceOverride -}}
{{- end }}

{{- define "amd-dra-driver.chart" -}}
{{- .Chart.Name }}-{{ .Chart.Version | replace "+" "_" }}
{{- end }}

{{/*
Standard labels applied to every top-level object.
*/}}
{{- define "amd-dra-driver.labels" -}}
helm.sh/chart: {{ include "amd-dra-driver.chart" . }}
app.kubernetes.io/version: {{ .Chart.AppVersion | quote }}
app.kubernetes.io/managed-by: {{ .Release.Service }}
{{ include "amd-dra-driver.templateLabels" . }}
{{- end }}

{{/*
Pod-template labels (smaller set).
*/}}
{{- define "amd-dra-driver.templateLabels" -}}
app.kubernetes.io/name: {{ include "amd-dra-driver.name" . }}
app.kubernetes.io/instance: {{ .Release.Name }}
{{- end }}

{{/*
resource.k8s.io API version the DeviceClasses are served under. "v1" on
k8s >= 1.34 clusters; v1beta1 otherwise. The reference templates the same
decision (deviceclass-gpu.yaml:2,12-14) — and only a v1 DeviceClass may
carry extendedResourceName.
*/}}
{{- define "amd-dra-driver.resourceApiVersion" -}}
{{- if eq .Values.resourceApiVersion "v1" -}}
resource.k8s.io/v1
{{- else -}}
resource.k8s.io/v1beta1
{{- end -}}
{{- end }}

{{/* Namespaces that run ComputeDomain daemon pods: the release namespace
     plus controller.additionalNamespaces (comma-separated), mirroring the
     controller's --additional-namespaces flag. */}}
{{- define "amd-dra-driver.cdNamespaces" -}}
{{- if .Values.controller.additionalNamespaces -}}
{{- printf "%s,%s" (include "amd-dra-driver.namespace" .) .Values.controller.additionalNamespaces -}}
{{- else -}}
{{- include "amd-dra-driver.namespace" . -}}
{{- end -}}
{{- end }}
