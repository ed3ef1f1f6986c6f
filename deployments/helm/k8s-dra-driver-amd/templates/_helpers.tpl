{{- define "k8s-dra-driver-amd.name" -}}
{{- default .Chart.Name .Values.nameOverride | trunc 63 | trimSuffix "-" -}}
{{- end -}}

{{- define "k8s-dra-driver-amd.namespace" -}}
{{- default .Release.Namespace .Values.namespaceOverride -}}
{{- end -}}

{{- define "k8s-dra-driver-amd.labels" -}}
app.kubernetes.io/name: {{ include "k8s-dra-driver-amd.name" . }}
app.kubernetes.io/instance: {{ .Release.Name }}
app.kubernetes.io/version: {{ .Chart.AppVersion | quote }}
app.kubernetes.io/managed-by: {{ .Release.Service }}
{{- end -}}

{{- define "k8s-dra-driver-amd.selectorLabels" -}}
{{- if .Values.selectorLabelsOverride -}}
{{ toYaml .Values.selectorLabelsOverride }}
{{- else -}}
app.kubernetes.io/name: {{ include "k8s-dra-driver-amd.name" . }}
app.kubernetes.io/instance: {{ .Release.Name }}
{{- end -}}
{{- end -}}

{{- define "k8s-dra-driver-amd.serviceAccountName" -}}
{{- if .Values.serviceAccount.create -}}
{{- default (include "k8s-dra-driver-amd.name" .) .Values.serviceAccount.name -}}
{{- else -}}
{{- default "default" .Values.serviceAccount.name -}}
{{- end -}}
{{- end -}}

{{- define "k8s-dra-driver-amd.image" -}}
{{- printf "%s:%s" .Values.image.repository (default .Chart.AppVersion .Values.image.tag) -}}
{{- end -}}

{{- define "k8s-dra-driver-amd.listHas" -}}
{{- $list := index . 0 -}}
{{- $item := index . 1 -}}
{{- if has $item $list -}}true{{- end -}}
{{- end -}}
