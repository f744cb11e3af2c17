{{- define "chart.fullname" -}}
{{ .Release.Name }}
{{- end }}

{{- define "chart.engineLabels" -}}
app.kubernetes.io/part-of: production-stack-amd
environment: engine
release: engine
{{- end }}

{{- define "chart.routerLabels" -}}
app.kubernetes.io/part-of: production-stack-amd
environment: router
release: router
{{- end }}
